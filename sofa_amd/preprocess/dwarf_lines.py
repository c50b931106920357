"""Minimal DWARF .debug_line reader: file:line for sampled addresses.

Round-1 symbolization was .symtab-only; with -g binaries this adds source
locations to flamegraphs and sample names.  Self-contained (no pyelftools in
the image): parses ELF section headers directly and executes the DWARF v2-v5
line-number programs.  Only the forms compilers actually emit for line
tables are handled; anything unrecognized aborts that unit silently — the
reader must never break symbolization.

Reference lineage: the original used `perf script`, which resolved lines via
libbfd when available; this is the native equivalent for the in-tree
sampler's offline pipeline.
"""

from __future__ import annotations

import bisect
import os
import struct
from typing import Dict, List, Optional, Tuple

# DW_FORMs needed for v5 directory/file tables
DW_FORM_string = 0x08
DW_FORM_strp = 0x0E
DW_FORM_udata = 0x0F
DW_FORM_line_strp = 0x1F
DW_FORM_data1 = 0x0B
DW_FORM_data2 = 0x05
DW_FORM_data4 = 0x06
DW_FORM_data8 = 0x07
DW_FORM_data16 = 0x1E
DW_FORM_block = 0x09

DW_LNCT_path = 1
DW_LNCT_directory_index = 2


def _elf_sections(data: bytes) -> Dict[str, Tuple[int, int]]:
    """name -> (offset, size) for ELF64 files."""
    if len(data) < 64 or data[:4] != b"\x7fELF" or data[4] != 2:
        return {}
    (e_shoff,) = struct.unpack_from("<Q", data, 0x28)
    (e_shentsize, e_shnum, e_shstrndx) = struct.unpack_from("<HHH", data, 0x3A)
    if e_shoff == 0 or e_shoff + e_shnum * e_shentsize > len(data):
        return {}
    raw = []
    for i in range(e_shnum):
        off = e_shoff + i * e_shentsize
        (sh_name,) = struct.unpack_from("<I", data, off)
        (sh_offset, sh_size) = struct.unpack_from("<QQ", data, off + 0x18)
        raw.append((sh_name, sh_offset, sh_size))
    if e_shstrndx >= len(raw):
        return {}
    str_off = raw[e_shstrndx][1]
    out = {}
    for sh_name, sh_offset, sh_size in raw:
        end = data.find(b"\0", str_off + sh_name)
        if end < 0:
            continue
        name = data[str_off + sh_name : end].decode("latin1")
        out[name] = (sh_offset, sh_size)
    return out


class _Reader:
    __slots__ = ("d", "pos", "end")

    def __init__(self, d: bytes, pos: int, end: int):
        self.d = d
        self.pos = pos
        self.end = end

    def u8(self) -> int:
        v = self.d[self.pos]
        self.pos += 1
        return v

    def u16(self) -> int:
        (v,) = struct.unpack_from("<H", self.d, self.pos)
        self.pos += 2
        return v

    def u32(self) -> int:
        (v,) = struct.unpack_from("<I", self.d, self.pos)
        self.pos += 4
        return v

    def u64(self) -> int:
        (v,) = struct.unpack_from("<Q", self.d, self.pos)
        self.pos += 8
        return v

    def uleb(self) -> int:
        result = shift = 0
        while True:
            b = self.d[self.pos]
            self.pos += 1
            result |= (b & 0x7F) << shift
            if not b & 0x80:
                return result
            shift += 7

    def sleb(self) -> int:
        result = shift = 0
        while True:
            b = self.d[self.pos]
            self.pos += 1
            result |= (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                if b & 0x40:
                    result -= 1 << shift
                return result

    def cstr(self) -> str:
        end = self.d.index(b"\0", self.pos)
        s = self.d[self.pos : end].decode("utf-8", "replace")
        self.pos = end + 1
        return s


def _strp(data: bytes, sec: Optional[Tuple[int, int]], off: int) -> str:
    if sec is None:
        return ""
    start = sec[0] + off
    end = data.find(b"\0", start)
    return data[start:end].decode("utf-8", "replace") if end >= 0 else ""


class LineTable:
    """addr -> (file, line) from every line program in an ELF's .debug_line."""

    def __init__(self, path: str):
        self.addrs: List[int] = []
        self.rows: List[Tuple[str, int]] = []  # parallel to addrs
        # cheap preflight: parse ONLY the section headers (a release
        # libtorch is ~2 GB; slurping it to discover there is no
        # .debug_line would double preprocess IO)
        try:
            if not self._has_debug_line(path):
                return
            with open(path, "rb") as f:
                data = f.read()
        except OSError:
            return
        secs = _elf_sections(data)
        dl = secs.get(".debug_line")
        if dl is None:
            return
        line_str = secs.get(".debug_line_str")
        debug_str = secs.get(".debug_str")
        pairs: List[Tuple[int, str, int]] = []  # (addr, file, line)
        r = _Reader(data, dl[0], dl[0] + dl[1])
        while r.pos < r.end:
            try:
                self._parse_unit(r, data, line_str, debug_str, pairs)
            except (IndexError, struct.error, ValueError):
                break
        pairs.sort()
        self.addrs = [p[0] for p in pairs]
        self.rows = [(p[1], p[2]) for p in pairs]

    @staticmethod
    def _has_debug_line(path: str) -> bool:
        with open(path, "rb") as f:
            hdr = f.read(64)
            if len(hdr) < 64 or hdr[:4] != b"\x7fELF" or hdr[4] != 2:
                return False
            (e_shoff,) = struct.unpack_from("<Q", hdr, 0x28)
            (e_shentsize, e_shnum, e_shstrndx) = struct.unpack_from("<HHH", hdr, 0x3A)
            if e_shoff == 0 or e_shnum == 0 or e_shstrndx >= e_shnum:
                return False
            f.seek(e_shoff)
            sh = f.read(e_shnum * e_shentsize)
            if len(sh) < e_shnum * e_shentsize:
                return False
            (str_off, str_size) = struct.unpack_from(
                "<QQ", sh, e_shstrndx * e_shentsize + 0x18
            )
            f.seek(str_off)
            shstr = f.read(str_size)
            return b".debug_line\0" in shstr

    # ---- one line-number program unit
    def _parse_unit(self, r: _Reader, data, line_str, debug_str, pairs) -> None:
        unit_len = r.u32()
        if unit_len in (0, 0xFFFFFFFF):  # 64-bit DWARF not emitted by gcc/clang here
            r.pos = r.end
            return
        unit_end = r.pos + unit_len
        version = r.u16()
        if version < 2 or version > 5:
            r.pos = unit_end
            return
        if version >= 5:
            r.u8()  # address_size
            r.u8()  # segment_selector_size
        header_len = r.u32()
        prog_start = r.pos + header_len
        min_inst = r.u8()
        if version >= 4:
            r.u8()  # max_ops_per_inst (VLIW only; 1 on x86/amdgcn)
        r.u8()  # default_is_stmt
        line_base = struct.unpack_from("<b", r.d, r.pos)[0]
        r.pos += 1
        line_range = r.u8()
        opcode_base = r.u8()
        std_lens = [r.u8() for _ in range(opcode_base - 1)]

        files: List[Tuple[str, int]] = []  # (name, dir_idx)
        dirs: List[str] = []
        if version >= 5:
            dirs, files = self._v5_tables(r, data, line_str, debug_str)
        else:
            dirs = [""]
            while True:
                s = r.cstr()
                if not s:
                    break
                dirs.append(s)
            files = [("", 0)]
            while True:
                name = r.cstr()
                if not name:
                    break
                d = r.uleb()
                r.uleb()  # mtime
                r.uleb()  # size
                files.append((name, d))

        def file_path(idx: int) -> str:
            if idx < 0 or idx >= len(files):
                return "?"
            name, d = files[idx]
            dirname = dirs[d] if 0 <= d < len(dirs) else ""
            return os.path.join(dirname, name) if dirname and not name.startswith("/") else name

        # ---- execute the state machine
        r.pos = prog_start
        address = 0
        file_idx = 1 if version < 5 else 1
        line = 1
        while r.pos < unit_end:
            op = r.u8()
            if op >= opcode_base:  # special opcode
                adj = op - opcode_base
                address += (adj // line_range) * min_inst
                line += line_base + (adj % line_range)
                pairs.append((address, file_path(file_idx), line))
            elif op == 0:  # extended
                ext_len = r.uleb()
                ext_end = r.pos + ext_len
                sub = r.u8() if ext_len else 0
                if sub == 1:  # DW_LNE_end_sequence
                    pairs.append((address, "", 0))  # sentinel: gap after
                    address, file_idx, line = 0, 1, 1
                elif sub == 2:  # DW_LNE_set_address
                    address = r.u64()
                # DW_LNE_define_file / vendor: skipped
                r.pos = ext_end
            elif op == 1:  # DW_LNS_copy
                pairs.append((address, file_path(file_idx), line))
            elif op == 2:  # advance_pc
                address += r.uleb() * min_inst
            elif op == 3:  # advance_line
                line += r.sleb()
            elif op == 4:  # set_file
                file_idx = r.uleb()
            elif op == 5:  # set_column
                r.uleb()
            elif op == 8:  # const_add_pc
                address += ((255 - opcode_base) // line_range) * min_inst
            elif op == 9:  # fixed_advance_pc
                address += r.u16()
            elif op in (6, 7, 10, 11):  # negate_stmt/basic_block/prologue/epilogue
                pass
            elif op == 12:  # set_isa
                r.uleb()
            else:  # unknown standard opcode: skip its operands
                n = std_lens[op - 1] if op - 1 < len(std_lens) else 0
                for _ in range(n):
                    r.uleb()
        r.pos = unit_end

    def _v5_tables(self, r: _Reader, data, line_str, debug_str):
        def read_entries():
            fmt_count = r.u8()
            fmts = [(r.uleb(), r.uleb()) for _ in range(fmt_count)]
            count = r.uleb()
            entries = []
            for _ in range(count):
                path, dir_idx = "", 0
                for content, form in fmts:
                    if form == DW_FORM_string:
                        val = r.cstr()
                    elif form == DW_FORM_line_strp:
                        val = _strp(data, line_str, r.u32())
                    elif form == DW_FORM_strp:
                        val = _strp(data, debug_str, r.u32())
                    elif form == DW_FORM_udata:
                        val = r.uleb()
                    elif form == DW_FORM_data1:
                        val = r.u8()
                    elif form == DW_FORM_data2:
                        val = r.u16()
                    elif form == DW_FORM_data4:
                        val = r.u32()
                    elif form == DW_FORM_data8:
                        val = r.u64()
                    elif form == DW_FORM_data16:
                        r.pos += 16
                        val = 0
                    elif form == DW_FORM_block:
                        n = r.uleb()
                        r.pos += n
                        val = 0
                    else:
                        raise ValueError(f"unhandled DW_FORM {form:#x}")
                    if content == DW_LNCT_path:
                        path = val if isinstance(val, str) else str(val)
                    elif content == DW_LNCT_directory_index:
                        dir_idx = int(val)
                entries.append((path, dir_idx))
            return entries

        dir_entries = read_entries()
        file_entries = read_entries()
        dirs = [p for p, _ in dir_entries]
        files = file_entries
        return dirs, files

    def lookup(self, file_addr: int) -> Optional[Tuple[str, int]]:
        """file-relative address -> (source file, line) or None."""
        if not self.addrs:
            return None
        i = bisect.bisect_right(self.addrs, file_addr) - 1
        if i < 0:
            return None
        f, ln = self.rows[i]
        if not f:  # end_sequence sentinel: address past the last range
            return None
        return f, ln
