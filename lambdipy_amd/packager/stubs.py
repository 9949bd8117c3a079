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

# init/teardown hooks the host library calls unconditionally at startup:
# these must succeed (return 0) rather than trap.
DEFAULT_BENIGN: Dict[str, Tuple[str, ...]] = {
    "libmagma.so": ("magma_init", "magma_finalize"),
}

_SYM_RE = re.compile(
    r"^\s*\d+:\s+[0-9a-f]+\s+(\d+)\s+(FUNC|OBJECT|NOTYPE|GNU_IFUNC|TLS)\s+"
    r"(GLOBAL|WEAK)\s+(DEFAULT|PROTECTED)\s+(\d+|[A-Z]+)\s+(\S+)")


def exported_symbols(so_path: str) -> List[Tuple[str, str, int, str]]:
    """(name, type, size, version) for dynamic symbols DEFINED by the DSO.

    version is '' for unversioned symbols; importers that reference
    name@VERSION need the stub to re-define the same version node."""
    out = subprocess.run(["readelf", "-W", "--dyn-syms", so_path],
                         capture_output=True, text=True, check=True).stdout
    syms = []
    for line in out.splitlines():
        m = _SYM_RE.match(line)
        if not m:
            continue
        size, typ, _bind, _vis, shndx, name = m.groups()
        if shndx in ("UND", "ABS"):
            continue  # imported / version-definition pseudo-symbols
        ver = ""
        if "@" in name:
            name, _, ver = name.partition("@")
            ver = ver.lstrip("@")
        syms.append((name, typ, int(size), ver))
    # dedupe (keep the default-versioned / larger entry)
    seen: Dict[str, Tuple[str, int, str]] = {}
    for name, typ, size, ver in syms:
        if name not in seen or seen[name][1] < size or (ver and not seen[name][2]):
            seen[name] = (typ, size, ver)
    return [(n, t, s, v) for n, (t, s, v) in sorted(seen.items())]


def soname_of(so_path: str) -> Optional[str]:
    out = subprocess.run(["readelf", "-d", so_path], capture_output=True,
                         text=True, check=True).stdout
    m = re.search(r"SONAME\)?\s+Library soname: \[([^\]]+)\]", out)
    return m.group(1) if m else None


def needed_of(so_path: str) -> List[str]:
    out = subprocess.run(["readelf", "-d", so_path], capture_output=True,
                         text=True, check=True).stdout
    return re.findall(r"NEEDED\)?\s+Shared library: \[([^\]]+)\]", out)


def make_stub(src_so: str, out_so: str, cc: str = "gcc",
              benign: Tuple[str, ...] = ()) -> Dict:
    """Generate a stub .so exporting src_so's symbol set under its SONAME.

    Functions trap when called (the layer is built on the premise that the
    dependency is never exercised); objects become zeroed storage of the
    original size so data references stay in-bounds. Symbols in `benign`
    return 0 instead of trapping — for init/finalize hooks the host calls
    unconditionally (e.g. torch's CUDA init calls magma_init()).
    """
    syms = exported_symbols(src_so)
    soname = soname_of(src_so) or os.path.basename(src_so)
    kept = [s for s in syms
            if re.fullmatch(r"[A-Za-z_$.][A-Za-z0-9_$.]*", s[0])]
    versions: Dict[str, List[str]] = {}
    with tempfile.TemporaryDirectory(prefix="lpx_stub_") as tmp:
        csrc = os.path.join(tmp, "stub.c")
        with open(csrc, "w") as f:
            f.write("__attribute__((noreturn)) static void lpx_stub_trap(void)"
                    "{__builtin_trap();}\n")
            for i, (name, typ, size, ver) in enumerate(kept):
                cname = f"lpx_s{i}"
                if typ == "OBJECT" or typ == "TLS":
                    sz = max(int(size), 1)
                    f.write(f'char {cname}[{sz}] __asm__("{name}") '
                            f'__attribute__((aligned(16)));\n')
                elif name in benign:
                    f.write(f'long {cname}(void) __asm__("{name}");\n'
                            f'long {cname}(void) {{ return 0; }}\n')
                else:
                    f.write(f'void {cname}(void) __asm__("{name}");\n'
                            f'void {cname}(void) {{ lpx_stub_trap(); }}\n')
                if ver:
                    versions.setdefault(ver, []).append(name)
        cmd = [cc, "-shared", "-fPIC", "-O0",
               "-fno-asynchronous-unwind-tables", f"-Wl,-soname,{soname}"]
        if versions:
            # replicate the original version nodes so name@VERSION refs
            # resolve (ld.so asserts otherwise)
            vs = os.path.join(tmp, "stub.ver")
            with open(vs, "w") as f:
                for ver, names in versions.items():
                    f.write(ver + " {\nglobal:\n")
                    for n in names:
                        f.write(f'  "{n}";\n')
                    f.write("};\n")
            cmd.append(f"-Wl,--version-script={vs}")
        cmd += ["-o", out_so, csrc]
        subprocess.run(cmd, check=True, capture_output=True)
    return {"src": src_so, "out": out_so, "soname": soname,
            "symbols": len(kept), "versions": sorted(versions),
            "old_size": os.path.getsize(src_so),
            "new_size": os.path.getsize(out_so)}
