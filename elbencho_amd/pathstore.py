"""Custom tree files: parse/write/scan of tree files and partitioning info.

Reference analogue: PathStore.{h,cpp} + treefile handling in
ProgArgs.cpp:2799-2932 (--treefile/--treescan/--treeroundup). Format:
    d <dirpath>
    f <size> <filepath>
Lines starting with '#' are comments. The reference optionally base64-encodes
paths with odd characters (header line "b64"); we accept and emit that too.

The actual per-rank split (non-shared round-robin, shared range slicing) is
done natively in the engine (csrc/engine.cpp customTreeFiles); this module
only loads/normalizes the tree.
"""

from __future__ import annotations

import base64
import os
from dataclasses import dataclass, field

B64_HEADER = "b64"


@dataclass
class CustomTree:
    dirs: list[str] = field(default_factory=list)
    files: list[tuple[str, int]] = field(default_factory=list)  # (relpath, size)

    def round_up(self, multiple: int) -> None:
        if multiple <= 0:
            return
        self.files = [(p, ((s + multiple - 1) // multiple) * multiple if s else 0)
                      for p, s in self.files]

    def total_file_bytes(self) -> int:
        return sum(s for _, s in self.files)

    def split_share(self, sharesize: int) -> tuple[list[tuple[str, int]], list[tuple[str, int]]]:
        """(non-shared, shared) sublists by --sharesize threshold."""
        if not sharesize:
            return list(self.files), []
        nonshared = [(p, s) for p, s in self.files if s < sharesize]
        shared = [(p, s) for p, s in self.files if s >= sharesize]
        return nonshared, shared


def parse_treefile(path: str) -> CustomTree:
    tree = CustomTree()
    b64 = False
    with open(path) as f:
        for lineno, raw in enumerate(f, 1):
            line = raw.rstrip("\n")
            if not line or line.startswith("#"):
                continue
            if lineno == 1 and line.strip() == B64_HEADER:
                b64 = True
                continue
            parts = line.split(" ", 2 if line[0] == "f" else 1)
            kind = parts[0]
            if kind == "d" and len(parts) == 2:
                tree.dirs.append(_decode(parts[1], b64))
            elif kind == "f" and len(parts) == 3:
                tree.files.append((_decode(parts[2], b64), int(parts[1])))
            else:
                raise ValueError(f"{path}:{lineno}: malformed tree file line: {line!r}")
    return tree


def _decode(p: str, b64: bool) -> str:
    return base64.b64decode(p).decode() if b64 else p.strip("/")


def _needs_b64(paths: list[str]) -> bool:
    return any("\n" in p or p != p.strip() for p in paths)


def write_treefile(tree: CustomTree, path: str) -> None:
    all_paths = tree.dirs + [p for p, _ in tree.files]
    b64 = _needs_b64(all_paths)
    with open(path, "w") as f:
        if b64:
            f.write(B64_HEADER + "\n")
        enc = (lambda p: base64.b64encode(p.encode()).decode()) if b64 else (lambda p: p)
        for d in tree.dirs:
            f.write(f"d {enc(d)}\n")
        for p, s in tree.files:
            f.write(f"f {s} {enc(p)}\n")


def scan_path(base: str) -> CustomTree:
    """--treescan: build a tree from an existing directory (reference
    FileTk custom-tree scanner)."""
    tree = CustomTree()
    base = base.rstrip("/")
    for root, dirs, files in os.walk(base):
        rel_root = os.path.relpath(root, base)
        for d in sorted(dirs):
            rel = d if rel_root == "." else f"{rel_root}/{d}"
            tree.dirs.append(rel)
        for fn in sorted(files):
            rel = fn if rel_root == "." else f"{rel_root}/{fn}"
            size = os.path.getsize(os.path.join(root, fn))
            tree.files.append((rel, size))
    # parents before children (mkdir order)
    tree.dirs.sort(key=lambda p: (p.count("/"), p))
    return tree
