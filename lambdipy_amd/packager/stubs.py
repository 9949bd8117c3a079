"""Stub generation for unused shared-library dependencies.

libtorch_hip.so hard-links (DT_NEEDED) several GPU math libraries the
lambdipy_amd inference path never calls — libmagma.so alone is 1.3 GB, and
MIOpen/rocsolver/rocsparse add another ~1.7 GB. The dynamic loader only
needs the DEFINED SYMBOLS to resolve, so a deployment layer can ship a
kilobyte-sized stub that exports the same symbol set (functions trap if
ever called; data symbols are zeroed storage of the original size) under
the same SONAME. Combined with fatbin.strip_fatbin this shrinks a working
gfx950 torch runtime from 11.5 GB to roughly a gigabyte.

Reference parity: lambdipy prunes wheel payloads to fit Lambda limits
(SURVEY.md §2.2 N6); stubbing is the MI355X-native extension of that idea
to the ROCm native-library layer.
"""

from __future__ import annotations

import os
import re
import subprocess
import tempfile
from typing import Dict, List, Optional, Tuple

_SYM_RE = re.compile(
    r"^\s*\d+:\s+[0-9a-f]+\s+(\d+)\s+(FUNC|OBJECT|NOTYPE|GNU_IFUNC|TLS)\s+"
    r"(GLOBAL|WEAK)\s+(DEFAULT|PROTECTED)\s+(\d+|[A-Z]+)\s+(\S+)")


def exported_symbols(so_path: str) -> List[Tuple[str, str, int]]:
    """(name, type, size) for dynamic symbols DEFINED by the DSO."""
    out = subprocess.run(["readelf", "-W", "--dyn-syms", so_path],
                         capture_output=True, text=True, check=True).stdout
    syms = []
    for line in out.splitlines():
        m = _SYM_RE.match(line)
        if not m:
            continue
        size, typ, _bind, _vis, shndx, name = m.groups()
        if shndx == "UND":
            continue  # imported, not exported
        if "@" in name:
            name = name.split("@")[0]
        syms.append((name, typ, int(size)))
    # dedupe (versioned aliases)
    seen = {}
    for name, typ, size in syms:
        if name not in seen or seen[name][1] < size:
            seen[name] = (typ, size)
    return [(n, t, s) for n, (t, s) in sorted(seen.items())]


def soname_of(so_path: str) -> Optional[str]:
    out = subprocess.run(["readelf", "-d", so_path], capture_output=True,
                         text=True, check=True).stdout
    m = re.search(r"SONAME\)?\s+Library soname: \[([^\]]+)\]", out)
    return m.group(1) if m else None


def needed_of(so_path: str) -> List[str]:
    out = subprocess.run(["readelf", "-d", so_path], capture_output=True,
                         text=True, check=True).stdout
    return re.findall(r"NEEDED\)?\s+Shared library: \[([^\]]+)\]", out)


def make_stub(src_so: str, out_so: str, cc: str = "gcc") -> Dict:
    """Generate a stub .so exporting src_so's symbol set under its SONAME.

    Functions trap when called (the layer is built on the premise that the
    dependency is never exercised); objects become zeroed storage of the
    original size so data references stay in-bounds.
    """
    syms = exported_symbols(src_so)
    soname = soname_of(src_so) or os.path.basename(src_so)
    with tempfile.TemporaryDirectory(prefix="lpx_stub_") as tmp:
        csrc = os.path.join(tmp, "stub.c")
        with open(csrc, "w") as f:
            f.write("__attribute__((noreturn)) static void lpx_stub_trap(void)"
                    "{__builtin_trap();}\n")
            for name, typ, size in syms:
                if not re.fullmatch(r"[A-Za-z_$.][A-Za-z0-9_$.]*", name):
                    continue  # skip unrepresentable names
                cname = name.replace(".", "_dot_").replace("$", "_dlr_")
                if typ == "OBJECT" or typ == "TLS":
                    sz = max(int(size), 1)
                    f.write(f'char {cname}[{sz}] __asm__("{name}") '
                            f'__attribute__((aligned(16)));\n')
                else:
                    f.write(f'void {cname}(void) __asm__("{name}");\n'
                            f'void {cname}(void) {{ lpx_stub_trap(); }}\n')
        subprocess.run([cc, "-shared", "-fPIC", "-O0", "-fno-asynchronous-unwind-tables",
                        f"-Wl,-soname,{soname}", "-o", out_so, csrc],
                       check=True, capture_output=True)
    return {"src": src_so, "out": out_so, "soname": soname,
            "symbols": len(syms), "old_size": os.path.getsize(src_so),
            "new_size": os.path.getsize(out_so)}
