"""Sanitizer build of the host engine + C API — counterpart of the
reference's ASan/UBSan CI jobs.  Builds libcimba with
-fsanitize=address,undefined using clang++ (hipcc's host compiler) and
runs the M/M/1 C tutorial under it; any leak/overflow/UB aborts."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLANGXX = "/opt/rocm/lib/llvm/bin/clang++"


@pytest.mark.skipif(not os.path.exists(CLANGXX), reason="no clang++")
def test_asan_ubsan_mm1(tmp_path):
    lib = str(tmp_path / "libcimba_asan.so")
    r = subprocess.run(
        [CLANGXX, "-std=c++17", "-O1", "-g", "-fPIC", "-shared",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "capi.cpp"),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", lib], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    obj = str(tmp_path / "mm1.o")
    r = subprocess.run(
        [CLANGXX, "-std=c11", "-xc", "-O1", "-g", "-c",
         "-fsanitize=address,undefined",
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", "mm1_capi.c"), "-o", obj],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    exe = str(tmp_path / "mm1_asan")
    r = subprocess.run(
        [CLANGXX, "-fsanitize=address,undefined",
         "-fno-sanitize-recover=all", obj, lib,
         f"-Wl,-rpath,{tmp_path}", "-lm", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    env = dict(os.environ,
               ASAN_OPTIONS="detect_leaks=1:abort_on_error=1",
               CIMBA_ASAN_TRIALS="1")
    out = subprocess.run([exe], capture_output=True, text=True, timeout=600,
                         env=env)
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-3000:])
    assert "Average system time" in out.stdout


@pytest.mark.skipif(not os.path.exists(CLANGXX), reason="no clang++")
def test_tsan_multithreaded_executive(tmp_path):
    """TSan over the multithreaded executive (reference runs TSan CI with
    fiber annotations; we have no fibers, so plain TSan applies)."""
    lib = str(tmp_path / "libcimba_tsan.so")
    r = subprocess.run(
        [CLANGXX, "-std=c++17", "-O1", "-g", "-fPIC", "-shared",
         "-fsanitize=thread",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "capi.cpp"),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", lib], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    obj = str(tmp_path / "mm1t.o")
    r = subprocess.run(
        [CLANGXX, "-std=c11", "-xc", "-O1", "-g", "-c",
         "-fsanitize=thread",
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", "mm1_capi.c"), "-o", obj],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    exe = str(tmp_path / "mm1_tsan")
    r = subprocess.run(
        [CLANGXX, "-fsanitize=thread", obj, lib,
         f"-Wl,-rpath,{tmp_path}", "-lm", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    env = dict(os.environ, TSAN_OPTIONS="halt_on_error=1")
    out = subprocess.run([exe], capture_output=True, text=True, timeout=900,
                         env=env)
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-3000:])


@pytest.mark.skipif(not os.path.exists(CLANGXX), reason="no clang++")
def test_asan_prodcons_stress(tmp_path):
    """ASan/UBSan over the randomized producer/consumer stress model —
    blocking queue paths, guard grants and timeout wakes under the
    multithreaded executive, with the sanitizer watching every
    engine-state access."""
    import re
    pcf = open(os.path.join(ROOT, "tests", "test_prodcons_fuzz.py")).read()
    HARNESS = re.search(r'HARNESS = r"""(.*?)"""', pcf, re.S).group(1)
    src = tmp_path / "pcfuzz_asan.cpp"
    src.write_text(HARNESS)
    exe = str(tmp_path / "pcfuzz_asan")
    r = subprocess.run(
        [CLANGXX, "-std=c++17", "-O1", "-g",
         "-fsanitize=address,undefined", "-fno-sanitize-recover=all",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    env = dict(os.environ, ASAN_OPTIONS="detect_leaks=1:abort_on_error=1")
    out = subprocess.run([exe, "7", "60"], capture_output=True, text=True,
                         timeout=900, env=env)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-3000:])
    assert "prodcons fuzz OK" in out.stdout


@pytest.mark.skipif(not os.path.exists(CLANGXX), reason="no clang++")
def test_tsan_prodcons_stress(tmp_path):
    """TSan over the same stress model running on all worker threads:
    each worker owns a private engine, so the only shared state is the
    work-claim counter and the report atomics — TSan proves it."""
    import re
    pcf = open(os.path.join(ROOT, "tests", "test_prodcons_fuzz.py")).read()
    HARNESS = re.search(r'HARNESS = r"""(.*?)"""', pcf, re.S).group(1)
    src = tmp_path / "pcfuzz_tsan.cpp"
    src.write_text(HARNESS)
    exe = str(tmp_path / "pcfuzz_tsan")
    r = subprocess.run(
        [CLANGXX, "-std=c++17", "-O1", "-g", "-fsanitize=thread",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    env = dict(os.environ, TSAN_OPTIONS="halt_on_error=1")
    out = subprocess.run([exe, "7", "60"], capture_output=True, text=True,
                         timeout=900, env=env)
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-3000:])
    assert "prodcons fuzz OK" in out.stdout


def test_asan_ubsan_spill_and_backmap(tmp_path):
    """The round-2 memory machinery under ASan+UBSan: heap/queue spill
    crossings (two-tier indexing, slab claim) and the handle back-map
    (open addressing, backward-shift deletion) — exactly the code where
    an off-by-one would corrupt silently in a release build."""
    import re

    spf = open(os.path.join(ROOT, "tests", "test_spill.py")).read()
    HARNESS = re.search(r'HARNESS = r"""(.*?)"""', spf, re.S).group(1)
    src = tmp_path / "spill_san.cpp"
    src.write_text(HARNESS)
    exe = str(tmp_path / "spill_san")
    r = subprocess.run(
        [CLANGXX, "-std=c++17", "-O1", "-g",
         "-fsanitize=address,undefined", "-fno-omit-frame-pointer",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    out = subprocess.run([exe], capture_output=True, text=True, timeout=900,
                         env={**os.environ,
                              "UBSAN_OPTIONS": "halt_on_error=1",
                              "ASAN_OPTIONS": "detect_leaks=1"})
    assert out.returncode == 0, (out.stdout[-1000:], out.stderr[-3000:])
    assert "spill suite OK" in out.stdout
