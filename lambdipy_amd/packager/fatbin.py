"""gfx950-only fat-binary stripping for HIP shared libraries.

ROCm ships every GPU library with clang-offload-bundle fat binaries for
~14 gfx architectures (libtorch_hip.so alone carries a 251 MB ``.hip_fatbin``
section; the full torch dist is 11.5 GB).  For an MI355X-only deployment
everything except the ``gfx950`` entries is dead weight.  This module
rewrites a DSO in place of that weight (reference parity: lambdipy trims
library payloads to fit Lambda's 250 MB layer limit — see SURVEY.md §2.2 N6
"strip/prune pass"; this is the MI355X-native equivalent for HIP fat
binaries, which the reference never had to deal with).

Mechanics (empirically verified against ROCm 7.2 libtorch_hip.so):
  - ``.hipFatBinSegment`` holds 24-byte wrappers {magic 0x48495046, version,
    ptr, reserved}; ``ptr`` is the vaddr of a bundle inside ``.hip_fatbin``,
    applied at load time through a R_X86_64_RELATIVE reloc whose addend
    equals that vaddr.
  - bundles are zstd-"CCOB" compressed offload bundles (or raw
    ``__CLANG_OFFLOAD_BUNDLE__`` archives); each carries one entry per arch
    plus an empty host stub.
  - we unbundle/rebundle each through ``clang-offload-bundler`` keeping only
    the host stub + requested archs, rewrite the section, repoint wrappers
    and reloc addends, and compact the file by splitting the containing
    PT_LOAD at the shrunk section's end (vaddrs of later sections are
    preserved; only file offsets shift, by a multiple of the page size, so
    vaddr/offset congruence holds).  One spare program-header slot is
    obtained by dropping PT_NOTE (not used at runtime).

The result dlopens and runs kernels exactly as before on gfx950 machines.

When the DSO has no PT_NOTE to reuse (e.g. libmagma.so), the section is
rewritten in place and its tail zero-filled: the file size is unchanged but
the zeros cost nothing after zip/zstd in the layer artifact. Measured on
the ROCm 7.2 torch dist (torch/lib): 5.43 -> 3.73 GB on disk
(libtorch_hip 423->200 MB, rocsolver 760->90, rocsparse 392->80,
rocrand 354->42, MIOpen 954->797, rocblas 55->30; magma zero-filled).
"""

from __future__ import annotations

import os
import shutil
import struct
import subprocess
import tempfile
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

BUNDLER = "/opt/rocm/lib/llvm/bin/clang-offload-bundler"
WRAPPER_MAGIC = 0x48495046  # 'FPIH'
RAW_MAGIC = b"__CLANG_OFFLOAD_BUNDLE__"
CCOB_MAGIC = b"CCOB"
PAGE = 4096
PT_LOAD, PT_NOTE = 1, 4
R_X86_64_64 = 1
R_X86_64_RELATIVE = 8


@dataclass
class Section:
    name: str
    sh_type: int
    flags: int
    addr: int
    offset: int
    size: int
    shdr_off: int  # file offset of this section header entry


@dataclass
class Phdr:
    p_type: int
    p_flags: int
    p_offset: int
    p_vaddr: int
    p_paddr: int
    p_filesz: int
    p_memsz: int
    p_align: int

    def pack(self) -> bytes:
        return struct.pack("<IIQQQQQQ", self.p_type, self.p_flags,
                           self.p_offset, self.p_vaddr, self.p_paddr,
                           self.p_filesz, self.p_memsz, self.p_align)


@dataclass
class Elf:
    data: bytearray
    e_shoff: int = 0
    e_phoff: int = 0
    e_phnum: int = 0
    e_shnum: int = 0
    e_shentsize: int = 0
    sections: List[Section] = field(default_factory=list)
    phdrs: List[Phdr] = field(default_factory=list)

    def section(self, name: str) -> Optional[Section]:
        for s in self.sections:
            if s.name == name:
                return s
        return None


def _parse_elf(data: bytearray) -> Elf:
    if data[:4] != b"\x7fELF" or data[4] != 2:
        raise ValueError("not a 64-bit ELF")
    e = Elf(data)
    (e_phoff, e_shoff) = struct.unpack("<QQ", data[0x20:0x30])
    (e_phentsize, e_phnum, e_shentsize, e_shnum, e_shstrndx) = struct.unpack(
        "<HHHHH", data[0x36:0x40])
    e.e_shoff, e.e_phoff = e_shoff, e_phoff
    e.e_phnum, e.e_shnum, e.e_shentsize = e_phnum, e_shnum, e_shentsize
    for i in range(e_phnum):
        off = e_phoff + i * e_phentsize
        vals = struct.unpack("<IIQQQQQQ", data[off:off + 56])
        e.phdrs.append(Phdr(*vals))
    # section headers
    shs = []
    for i in range(e_shnum):
        off = e_shoff + i * e_shentsize
        (nameoff, sh_type, flags, addr, offset, size) = struct.unpack(
            "<IIQQQQ", data[off:off + 40])
        shs.append((nameoff, sh_type, flags, addr, offset, size, off))
    stroff = shs[e_shstrndx][4]
    for (nameoff, sh_type, flags, addr, offset, size, off) in shs:
        end = data.index(b"\x00", stroff + nameoff)
        name = bytes(data[stroff + nameoff:end]).decode()
        e.sections.append(Section(name, sh_type, flags, addr, offset, size, off))
    return e


def _vaddr_to_off(e: Elf, vaddr: int) -> int:
    for p in e.phdrs:
        if p.p_type == PT_LOAD and p.p_vaddr <= vaddr < p.p_vaddr + p.p_filesz:
            return p.p_offset + (vaddr - p.p_vaddr)
    raise ValueError(f"vaddr {vaddr:#x} not in any PT_LOAD")


def _bundle_extent(data: bytes, off: int) -> int:
    """Byte length of the bundle starting at `off` (CCOB or raw)."""
    if data[off:off + 4] == CCOB_MAGIC:
        ver, _method = struct.unpack("<HH", data[off + 4:off + 8])
        if ver in (1, 2):
            (total,) = struct.unpack("<I", data[off + 8:off + 12])
        elif ver == 3:
            (total,) = struct.unpack("<Q", data[off + 8:off + 16])
        else:
            raise ValueError(f"unknown CCOB version {ver}")
        return total
    if data[off:off + 24] == RAW_MAGIC:
        p = off + 24
        (n,) = struct.unpack("<Q", data[p:p + 8])
        p += 8
        end = off + 32
        for _ in range(n):
            o, s, l = struct.unpack("<QQQ", data[p:p + 24])
            p += 24 + l
            end = max(end, off + o + s)
        return max(end, p) - off
    raise ValueError(f"no bundle magic at {off:#x}")


def _list_targets(path: str, bundler: str) -> List[str]:
    out = subprocess.run([bundler, "--type=o", "--list", f"--input={path}"],
                         capture_output=True, text=True, check=True)
    return [t for t in out.stdout.split() if t]


def _rebundle(blob: bytes, keep_archs: Tuple[str, ...], tmp: str,
              bundler: str, tag: str, level: int) -> Optional[bytes]:
    """Rebundle keeping host + keep_archs. None => keep original bytes."""
    src = os.path.join(tmp, f"in{tag}.bundle")
    with open(src, "wb") as f:
        f.write(blob)
    targets = _list_targets(src, bundler)
    host = [t for t in targets if t.startswith("host-")]
    keep = [t for t in targets
            if any(t.endswith("-" + a) or t.endswith("--" + a) for a in keep_archs)]
    if not keep or len(keep) + len(host) == len(targets):
        return None  # nothing to strip (or target arch absent: leave intact)
    sel = host[:1] + keep
    outs = [os.path.join(tmp, f"t{tag}_{i}.bin") for i in range(len(sel))]
    cmd = [bundler, "--type=o", "--unbundle", f"--input={src}",
           "--targets=" + ",".join(sel)] + [f"--output={o}" for o in outs]
    subprocess.run(cmd, check=True, capture_output=True)
    dst = os.path.join(tmp, f"out{tag}.bundle")
    cmd = [bundler, "--type=o", "--targets=" + ",".join(sel),
           f"--output={dst}", "-compress", f"-compression-level={level}"]
    cmd += [f"--input={o}" for o in outs]
    subprocess.run(cmd, check=True, capture_output=True)
    with open(dst, "rb") as f:
        new = f.read()
    for o in outs + [src, dst]:
        os.unlink(o)
    return new if len(new) < len(blob) else blob


def strip_fatbin(src: str, dst: Optional[str] = None,
                 archs: Tuple[str, ...] = ("gfx950",),
                 bundler: str = BUNDLER, verbose: bool = False,
                 level: int = 9, jobs: int = 8) -> Dict:
    """Strip non-`archs` GPU code from a HIP DSO. Returns a stats dict.

    Writes to `dst` (default: in-place via temp + rename). No-op (still
    copies) when the DSO has no .hip_fatbin section.
    """
    with open(src, "rb") as f:
        data = bytearray(f.read())
    e = _parse_elf(data)
    fat = e.section(".hip_fatbin")
    seg = e.section(".hipFatBinSegment")
    stats = {"src": src, "old_size": len(data), "bundles": 0, "stripped": 0}
    if fat is None or seg is None or fat.size == 0:
        if dst and dst != src:
            shutil.copyfile(src, dst)
        stats["new_size"] = len(data)
        return stats

    # 1. wrappers -> bundle vaddrs. The in-file ptr field may be zero (set
    # only at load time), so resolve via the reloc targeting it: either a
    # RELATIVE addend (libtorch_hip style) or an R_X86_64_64 against a
    # dynsym whose st_value is the bundle vaddr (librccl style).
    rela = e.section(".rela.dyn")
    dynsym = e.section(".dynsym")
    rel_by_off: Dict[int, int] = {}
    sym_by_off: Dict[int, Tuple[int, int]] = {}  # ptr-field vaddr -> (sym, add)
    if rela is not None:
        for off in range(rela.offset, rela.offset + rela.size, 24):
            r_offset, r_info, r_addend = struct.unpack("<QQQ", data[off:off + 24])
            rtype = r_info & 0xffffffff
            if rtype == R_X86_64_RELATIVE:
                rel_by_off[r_offset] = r_addend
            elif rtype == R_X86_64_64:
                sym_by_off[r_offset] = (r_info >> 32, r_addend)

    def sym_value(idx: int) -> int:
        off = dynsym.offset + idx * 24
        return struct.unpack("<Q", data[off + 8:off + 16])[0]

    def set_sym_value(idx: int, v: int) -> None:
        struct.pack_into("<Q", data, dynsym.offset + idx * 24 + 8, v)

    seg_off = _vaddr_to_off(e, seg.addr)
    wrappers = []  # (file_off_of_ptr_field, bundle_vaddr)
    wrapper_syms: Dict[int, int] = {}  # bundle vaddr -> dynsym index
    for i in range(seg.size // 24):
        p = seg_off + i * 24
        magic, _ver, ptr = struct.unpack("<IIQ", data[p:p + 16])
        if magic != WRAPPER_MAGIC:
            raise ValueError(f"bad wrapper magic at {p:#x}")
        field_va = seg.addr + i * 24 + 8
        if field_va in rel_by_off:
            ptr = rel_by_off[field_va]
        elif field_va in sym_by_off and dynsym is not None:
            sidx, add = sym_by_off[field_va]
            ptr = sym_value(sidx) + add
            wrapper_syms[ptr - add] = sidx
        wrappers.append((p + 8, ptr))
    stats["bundles"] = len(wrappers)

    # 2. rebundle each (subprocess-bound -> thread pool)
    fat_off = _vaddr_to_off(e, fat.addr)
    new_blobs: Dict[int, bytes] = {}  # old vaddr -> new bytes
    uniq = []
    for _, va in wrappers:
        if va not in new_blobs:
            new_blobs[va] = b""
            uniq.append(va)
    raw = bytes(data)
    with tempfile.TemporaryDirectory(prefix="lpx_fatbin_") as tmp:
        def work(va):
            off = fat_off + (va - fat.addr)
            blob = raw[off:off + _bundle_extent(raw, off)]
            return va, blob, _rebundle(blob, archs, tmp, bundler,
                                       f"{va:x}", level)

        from concurrent.futures import ThreadPoolExecutor
        with ThreadPoolExecutor(max_workers=jobs) as pool:
            for va, blob, new in pool.map(work, uniq):
                if new is None:
                    new = blob
                else:
                    stats["stripped"] += 1
                new_blobs[va] = new

    # 3. lay out the new section (64-B aligned bundles)
    layout: Dict[int, int] = {}
    cur = 0
    out = bytearray()
    for _, va in wrappers:
        if va in layout:
            continue
        cur = (cur + 63) & ~63
        out.extend(b"\x00" * (cur - len(out)))
        layout[va] = fat.addr + cur
        out.extend(new_blobs[va])
        cur = len(out)
    new_size = len(out)
    delta = ((fat.size - new_size) // PAGE) * PAGE
    keep = fat.size - delta  # bytes of the old section span we keep on disk
    out.extend(b"\x00" * (keep - new_size))

    # 4. patch wrapper ptr fields
    for ptr_field_off, va in wrappers:
        struct.pack_into("<Q", data, ptr_field_off, layout[va])

    # 5. patch RELATIVE reloc addends that point at old bundle starts, and
    # dynsym st_values for symbol-addressed bundles (R_X86_64_64 style)
    if rela is not None:
        for off in range(rela.offset, rela.offset + rela.size, 24):
            r_offset, r_info, r_addend = struct.unpack("<QQQ", data[off:off + 24])
            if (r_info & 0xffffffff) == R_X86_64_RELATIVE and r_addend in layout:
                struct.pack_into("<Q", data, off + 16, layout[r_addend])
    for old_va, sidx in wrapper_syms.items():
        if old_va in layout:
            set_sym_value(sidx, layout[old_va])

    # 6. splice: [fat_off, fat_off+keep) := out; drop the next `delta` bytes.
    # The segment split needs a spare phdr slot (we drop one PT_NOTE). If
    # there is none, fall back to in-place rewrite (no file-size reclaim).
    if delta and not any(p.p_type == PT_NOTE for p in e.phdrs):
        delta = 0
        keep = fat.size
        out.extend(b"\x00" * (keep - len(out)))
    if delta:
        # split the containing PT_LOAD
        li = next(i for i, p in enumerate(e.phdrs)
                  if p.p_type == PT_LOAD and
                  p.p_vaddr <= fat.addr < p.p_vaddr + p.p_filesz)
        L = e.phdrs[li]
        end_va = L.p_vaddr + L.p_filesz
        a = Phdr(PT_LOAD, L.p_flags, L.p_offset, L.p_vaddr, L.p_paddr,
                 (fat.addr + keep) - L.p_vaddr, (fat.addr + keep) - L.p_vaddr,
                 L.p_align)
        b_va = fat.addr + fat.size
        b = Phdr(PT_LOAD, L.p_flags, fat_off + keep, b_va, b_va,
                 end_va - b_va, end_va - b_va, L.p_align)
        assert (b.p_vaddr - b.p_offset) % PAGE == 0
        phdrs = list(e.phdrs)
        phdrs.remove(next(p for p in phdrs if p.p_type == PT_NOTE))
        phdrs[phdrs.index(L)] = a
        phdrs.append(b)
        loads = sorted([p for p in phdrs if p.p_type == PT_LOAD],
                       key=lambda p: p.p_vaddr)
        others = [p for p in phdrs if p.p_type != PT_LOAD]
        new_phdrs = loads + others
        for p in new_phdrs:
            if p is not b and p.p_offset >= fat_off + fat.size:
                p.p_offset -= delta
        for i, p in enumerate(new_phdrs):
            data[e.e_phoff + i * 56:e.e_phoff + (i + 1) * 56] = p.pack()
        # section header table: shift offsets past the cut; resize .hip_fatbin
        for s in e.sections:
            new_off = s.offset - delta if s.offset >= fat_off + fat.size else s.offset
            new_sz = new_size if s.shdr_off == fat.shdr_off else s.size
            struct.pack_into("<QQ", data, s.shdr_off + 24, new_off, new_sz)
        if e.e_shoff >= fat_off + fat.size:
            struct.pack_into("<Q", data, 0x28, e.e_shoff - delta)
        final = bytes(data[:fat_off]) + bytes(out) + bytes(data[fat_off + fat.size:])
    else:
        struct.pack_into("<Q", data, fat.shdr_off + 32, new_size)
        final = bytes(data[:fat_off]) + bytes(out) + bytes(data[fat_off + fat.size:])

    stats["new_size"] = len(final)
    stats["fatbin_old"] = fat.size
    stats["fatbin_new"] = new_size
    target = dst or src
    tmp_path = target + ".lpxtmp"
    with open(tmp_path, "wb") as f:
        f.write(final)
    os.chmod(tmp_path, 0o755)
    os.replace(tmp_path, target)
    if verbose:
        print(f"{src}: {stats['old_size']/1e6:.1f} -> {stats['new_size']/1e6:.1f} MB"
              f" (fatbin {fat.size/1e6:.1f} -> {new_size/1e6:.1f} MB,"
              f" {stats['stripped']}/{stats['bundles']} bundles stripped)")
    return stats
