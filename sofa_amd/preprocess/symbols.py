"""Offline symbolization for CPU samples.

Replaces the reference's reliance on `perf script` symbol resolution +
cxxfilt (bin/sofa_preprocess.py:405-414,1814-1816):

* per-pid address spaces are rebuilt from the sampler's MMAP records;
* user-space symbols come from a minimal ELF64 .symtab/.dynsym reader
  (no pyelftools in the image);
* kernel symbols come from the recorded /proc/kallsyms snapshot;
* demangling goes through libstdc++ __cxa_demangle via ctypes.

Resolution is vectorized: sorted np.searchsorted over mapping starts and over
per-DSO symbol tables.
"""

from __future__ import annotations

import bisect
import ctypes
import os
import struct
from functools import lru_cache
from typing import Dict, List, Optional, Tuple

import numpy as np

# ------------------------------------------------------------------ demangle

_libcxx = None


def _get_libcxx():
    global _libcxx
    if _libcxx is None:
        try:
            _libcxx = ctypes.CDLL("libstdc++.so.6")
            _libcxx.__cxa_demangle.restype = ctypes.c_void_p
            _libcxx.__cxa_demangle.argtypes = [
                ctypes.c_char_p,
                ctypes.c_void_p,
                ctypes.c_void_p,
                ctypes.POINTER(ctypes.c_int),
            ]
        except OSError:
            _libcxx = False
    return _libcxx


_libc = ctypes.CDLL(None, use_errno=True)


@lru_cache(maxsize=65536)
def demangle(name: str) -> str:
    if not name.startswith("_Z"):
        return name
    lib = _get_libcxx()
    if not lib:
        return name
    status = ctypes.c_int(0)
    ptr = lib.__cxa_demangle(name.encode(), None, None, ctypes.byref(status))
    if status.value == 0 and ptr:
        try:
            return ctypes.cast(ptr, ctypes.c_char_p).value.decode("utf-8", "replace")
        finally:
            _libc.free(ctypes.c_void_p(ptr))
    return name


# ------------------------------------------------------------------ ELF read


def read_elf_symbols(path: str) -> List[Tuple[int, int, str]]:
    """Return sorted [(addr, size, name)] of FUNC symbols from an ELF64 file."""
    try:
        with open(path, "rb") as f:
            data = f.read()
    except OSError:
        return []
    if len(data) < 64 or data[:4] != b"\x7fELF" or data[4] != 2:  # ELF64 only
        return []
    (e_shoff,) = struct.unpack_from("<Q", data, 0x28)
    (e_shentsize, e_shnum) = struct.unpack_from("<HH", data, 0x3A)
    if e_shoff == 0 or e_shoff + e_shnum * e_shentsize > len(data):
        return []
    sections = []
    for i in range(e_shnum):
        off = e_shoff + i * e_shentsize
        (sh_name, sh_type) = struct.unpack_from("<II", data, off)
        (sh_offset, sh_size) = struct.unpack_from("<QQ", data, off + 0x18)
        (sh_link,) = struct.unpack_from("<I", data, off + 0x28)
        (sh_entsize,) = struct.unpack_from("<Q", data, off + 0x38)
        sections.append((sh_type, sh_offset, sh_size, sh_link, sh_entsize))
    out: List[Tuple[int, int, str]] = []
    for sh_type, sh_offset, sh_size, sh_link, sh_entsize in sections:
        if sh_type not in (2, 11):  # SHT_SYMTAB, SHT_DYNSYM
            continue
        if sh_entsize == 0 or sh_link >= len(sections):
            continue
        str_off, str_size = sections[sh_link][1], sections[sh_link][2]
        strtab = data[str_off : str_off + str_size]
        n_syms = sh_size // sh_entsize
        for j in range(n_syms):
            off = sh_offset + j * sh_entsize
            (st_name, st_info) = struct.unpack_from("<IB", data, off)
            (st_value, st_size) = struct.unpack_from("<QQ", data, off + 8)
            if (st_info & 0xF) != 2:  # STT_FUNC
                continue
            if st_value == 0:
                continue
            end = strtab.find(b"\0", st_name)
            if end < 0:
                continue
            name = strtab[st_name:end].decode("utf-8", "replace")
            if name:
                out.append((st_value, st_size, name))
    out.sort()
    return out


def elf_is_pie(path: str) -> bool:
    """ET_DYN (shared object / PIE) => mapped addresses are base + st_value."""
    try:
        with open(path, "rb") as f:
            hdr = f.read(20)
        return len(hdr) >= 18 and hdr[:4] == b"\x7fELF" and struct.unpack_from("<H", hdr, 16)[0] == 3
    except OSError:
        return False


class DsoSymbols:
    def __init__(self, path: str):
        self.path = path
        self._lines = None  # lazy DWARF .debug_line table (None = not tried)
        syms = read_elf_symbols(path)
        self.addrs = np.array([s[0] for s in syms], dtype=np.uint64)
        self.names = [s[2] for s in syms]
        self.sizes = np.array([s[1] for s in syms], dtype=np.uint64)
        self.pie = elf_is_pie(path)

    def line_for(self, file_addr: int):
        """(source file basename, line) via .debug_line, or None."""
        if self._lines is None:
            from .dwarf_lines import LineTable

            self._lines = LineTable(self.path)
        hit = self._lines.lookup(file_addr)
        if hit is None:
            return None
        f, ln = hit
        import os as _os

        return _os.path.basename(f), ln

    def resolve(self, file_addr: int) -> Optional[str]:
        if len(self.addrs) == 0:
            return None
        i = int(np.searchsorted(self.addrs, file_addr, side="right")) - 1
        if i < 0:
            return None
        base = int(self.addrs[i])
        size = int(self.sizes[i])
        if size > 0 and file_addr >= base + size:
            return None
        if size == 0 and file_addr - base > (1 << 20):
            return None
        return self.names[i]


class KernelSymbols:
    def __init__(self, kallsyms_path: str):
        addrs: List[int] = []
        names: List[str] = []
        try:
            with open(kallsyms_path) as f:
                for line in f:
                    parts = line.split()
                    if len(parts) < 3:
                        continue
                    try:
                        addr = int(parts[0], 16)
                    except ValueError:
                        continue
                    if parts[1].lower() not in ("t", "w"):
                        continue
                    if addr == 0:
                        continue
                    addrs.append(addr)
                    names.append(parts[2])
        except OSError:
            pass
        order = np.argsort(np.array(addrs, dtype=np.uint64)) if addrs else []
        self.addrs = np.array([addrs[i] for i in order], dtype=np.uint64) if addrs else np.empty(0, np.uint64)
        self.names = [names[i] for i in order] if addrs else []

    def resolve(self, addr: int) -> Optional[str]:
        if len(self.addrs) == 0:
            return None
        i = int(np.searchsorted(self.addrs, addr, side="right")) - 1
        if i < 0:
            return None
        return self.names[i]


class Symbolizer:
    """Per-pid address space from sampler MMAP records + lazy DSO loading."""

    def __init__(
        self,
        mmaps: Dict[int, List[Tuple[int, int, int, str]]],
        kallsyms: str = "",
        container_root: str = "",
    ):
        # pid -> sorted list of (start, end, pgoff, path)
        self.spaces: Dict[int, List[Tuple[int, int, int, str]]] = {}
        for pid, maps in mmaps.items():
            entries = sorted((a, a + ln, off, name) for (a, ln, off, name) in maps)
            self.spaces[pid] = entries
        self._dsos: Dict[str, DsoSymbols] = {}
        self.ksyms = KernelSymbols(kallsyms) if kallsyms else None
        # in-container DSO paths resolve under the container overlayfs root
        # recorded by record.docker_target (reference used a bindfs symfs,
        # bin/sofa_preprocess.py:396-414)
        self.container_root = container_root.rstrip("/")

    def _dso(self, path: str) -> DsoSymbols:
        if path not in self._dsos:
            self._dsos[path] = DsoSymbols(path)
        return self._dsos[path]

    def resolve(self, pid: int, ip: int, kernel: bool = False) -> Tuple[str, str]:
        """Return (symbol, dso_basename)."""
        if kernel:
            if self.ksyms:
                s = self.ksyms.resolve(ip)
                if s:
                    return (s, "[kernel]")
            return ("0x%x" % ip, "[kernel]")
        space = self.spaces.get(pid)
        if space:
            starts = [e[0] for e in space]
            i = bisect.bisect_right(starts, ip) - 1
            if i >= 0:
                start, end, pgoff, path = space[i]
                if not os.path.isfile(path) and self.container_root:
                    alt = self.container_root + path
                    if os.path.isfile(alt):
                        path = alt
                if ip < end and os.path.isfile(path):
                    dso = self._dso(path)
                    file_addr = (ip - start + pgoff) if dso.pie else ip
                    sym = dso.resolve(file_addr)
                    base = os.path.basename(path)
                    if sym:
                        out = demangle(sym)
                        loc = dso.line_for(file_addr)
                        if loc is not None:
                            out = "%s (%s:%d)" % (out, loc[0], loc[1])
                        return (out, base)
                    return ("0x%x" % ip, base)
                elif ip < end:
                    return ("0x%x" % ip, os.path.basename(path))
        return ("0x%x" % ip, "??")
