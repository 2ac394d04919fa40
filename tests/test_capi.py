"""C embedding surface: compile and run a standalone C program against
libparsec_amd.so (the reference is consumed as a C library; capi.cpp is
that surface here)."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_c_embedding(tmp_path):
    exe = str(tmp_path / "c_embed")
    r = subprocess.run(
        ["gcc", "-O2", os.path.join(REPO, "examples", "c_embed.c"),
         "-o", exe, "-L" + os.path.join(REPO, "parsec_amd"),
         "-l:libparsec_amd.so",
         "-Wl,-rpath," + os.path.join(REPO, "parsec_amd")],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    r2 = subprocess.run([exe], capture_output=True, text=True, timeout=120)
    assert r2.returncode == 0 and "C_EMBED_OK" in r2.stdout, \
        r2.stdout + r2.stderr
