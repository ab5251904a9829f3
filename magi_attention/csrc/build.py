"""In-tree build of the MI355X native library (libmagi_ffa.so).

Invoked by __graft_entry__.build() and by `python -m magi_attention.csrc.build`.
Uses hipcc directly (no torch headers needed: the boundary is a pure C ABI,
see include/magi_ffa.h). The built .so lives in magi_attention/_libs/ so it
travels with the repo snapshot to the GPU box.
"""
from __future__ import annotations

import hashlib
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent
PKG = CSRC.parent
LIBDIR = PKG / "_libs"
LIB = LIBDIR / "libmagi_ffa.so"
STAMP = LIBDIR / ".build_stamp"

SOURCES = [
    CSRC / "ffa_fwd.hip",
    CSRC / "ffa_index.hip",
    CSRC / "ffa_bwd.hip",
    CSRC / "range_ops.hip",
    CSRC / "ffa_fwd_fp8.hip",
    CSRC / "ext_utils.hip",
    CSRC / "grpcoll.hip",
]

HIPCC = "hipcc"
FLAGS = [
    "--offload-arch=gfx950",
    "-O3",
    "-std=c++17",
    "-shared",
    "-fPIC",
    "-Wno-unused-result",
    "-munsafe-fp-atomics",
]


def _source_digest() -> str:
    h = hashlib.sha256()
    for s in sorted(SOURCES) + [PKG.parent / "include" / "magi_ffa.h"]:
        if s.exists():
            h.update(s.read_bytes())
    h.update(" ".join(FLAGS).encode())
    return h.hexdigest()


def build(force: bool = False, verbose: bool = True) -> Path:
    LIBDIR.mkdir(exist_ok=True)
    (LIBDIR / "__init__.py").touch()
    digest = _source_digest()
    if not force and LIB.exists() and STAMP.exists() and STAMP.read_text() == digest:
        return LIB
    srcs = [str(s) for s in SOURCES if s.exists()]
    cmd = [HIPCC, *FLAGS, *srcs, "-o", str(LIB)]
    if verbose:
        print("[magi_attention build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    STAMP.write_text(digest)
    return LIB


if __name__ == "__main__":
    build(force="--force" in sys.argv)
