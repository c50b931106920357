"""Tests for the native CPU sampler + SCS parser + symbolizer (CPU-only)."""

import os
import signal
import subprocess
import sys
import time

import numpy as np
import pytest

from sofa_amd.preprocess.scs import SAMPLE_DTYPE, parse_scs
from sofa_amd.preprocess import cpu as cpu_mod
from sofa_amd.preprocess.symbols import demangle, read_elf_symbols


def _sampler_path(repo_root):
    return os.path.join(repo_root, "sofa_amd", "native", "bin", "sofa-cpusampler")


def test_sampler_records_busy_process(tmp_path, repo_root, native_built):
    out = tmp_path / "t.scs"
    busy = subprocess.Popen(
        [sys.executable, "-c", "import time\nt=time.time()\nx=0\nwhile time.time()-t<2.0: x+=1"]
    )
    samp = subprocess.Popen([_sampler_path(repo_root), "-o", str(out), "-F", "99", "-p", str(busy.pid)])
    busy.wait()
    time.sleep(0.3)
    samp.send_signal(signal.SIGTERM)
    samp.wait(timeout=5)

    scs = parse_scs(str(out))
    # ~99 Hz over ~2s of a single busy thread
    assert len(scs.samples) > 100, f"too few samples: {len(scs.samples)}"
    assert scs.sample_freq == 99
    assert scs.realtime_ns > 0 and scs.monotonic_raw_ns > 0
    # sample timestamps are CLOCK_MONOTONIC_RAW and monotone non-decreasing per cpu
    s = scs.samples
    assert (s["period"] > 0).all()
    assert s["time_ns"].max() > scs.monotonic_raw_ns
    # the busy pid dominates
    pids, counts = np.unique(s["pid"], return_counts=True)
    assert pids[np.argmax(counts)] == busy.pid
    # mmap snapshot captured the python binary mapping
    assert busy.pid in scs.mmaps
    assert any("python" in m[3] or "libc" in m[3] for m in scs.mmaps[busy.pid])


def test_scs_to_cputrace(tmp_path, repo_root, native_built):
    out = tmp_path / "t.scs"
    busy = subprocess.Popen(
        [sys.executable, "-c", "import time\nt=time.time()\nx=0\nwhile time.time()-t<1.5: x+=1"]
    )
    samp = subprocess.Popen([_sampler_path(repo_root), "-o", str(out), "-F", "99", "-p", str(busy.pid)])
    busy.wait()
    samp.send_signal(signal.SIGTERM)
    samp.wait(timeout=5)

    scs = parse_scs(str(out))
    df = cpu_mod.scs_to_cputrace(scs, None, symbolize=True)
    assert len(df) == len(scs.samples)
    assert (df["duration"] > 0).all()
    # event = log10(ip)
    assert (df["event"] > 9).all()  # userspace addrs ~2^47 -> log10 ~14
    # at least some samples resolve to a real symbol or dso
    assert (~df["name"].str.startswith("0x")).any() or df["name"].str.contains("@").all()


def test_read_elf_symbols_on_libc():
    import ctypes.util

    # libc always present; its dynsym has FUNC symbols
    for cand in ["/usr/lib/x86_64-linux-gnu/libc.so.6", "/lib/x86_64-linux-gnu/libc.so.6"]:
        if os.path.exists(cand):
            syms = read_elf_symbols(cand)
            names = {s[2] for s in syms}
            assert "malloc" in names
            return
    pytest.skip("no libc found")


def test_demangle():
    assert demangle("_Z3addii") == "add(int, int)"
    assert demangle("main") == "main"
    assert demangle("not_mangled") == "not_mangled"


def test_dwarf_line_table(tmp_path):
    """DWARF .debug_line reader: file:line for a -g-compiled function."""
    import subprocess

    from sofa_amd.preprocess.dwarf_lines import LineTable
    from sofa_amd.preprocess.symbols import read_elf_symbols

    csrc = tmp_path / "hot.c"
    csrc.write_text(
        "int helper(int x) { return x * 3 + 1; }\n"
        "int work(int n) {\n"
        "  int s = 0;\n"
        "  for (int i = 0; i < n; i++) s += helper(i);\n"
        "  return s;\n"
        "}\n"
        "int main(void) { return work(10) & 0; }\n"
    )
    exe = tmp_path / "hot"
    subprocess.run(
        ["gcc", "-g", "-O0", "-no-pie", str(csrc), "-o", str(exe)],
        check=True, capture_output=True,
    )
    lt = LineTable(str(exe))
    assert lt.addrs, "no line rows parsed"
    syms = {name: addr for addr, _, name in read_elf_symbols(str(exe))}
    assert "work" in syms
    hit = lt.lookup(syms["work"] + 8)
    assert hit is not None
    fname, line = hit
    assert fname.endswith("hot.c")
    assert 2 <= line <= 6, line


def test_symbolizer_appends_file_line(tmp_path):
    """End-to-end: a sampled IP inside a -g binary resolves to
    'func (file.c:N)'."""
    import subprocess

    from sofa_amd.preprocess.symbols import Symbolizer, read_elf_symbols

    csrc = tmp_path / "app.c"
    csrc.write_text("int spin(int n){int s=0;for(int i=0;i<n;i++)s+=i;return s;}\n"
                    "int main(void){return spin(5)&0;}\n")
    exe = tmp_path / "app"
    subprocess.run(["gcc", "-g", "-O0", "-no-pie", str(csrc), "-o", str(exe)],
                   check=True, capture_output=True)
    syms = {name: addr for addr, _, name in read_elf_symbols(str(exe))}
    ip = syms["spin"] + 4
    s = Symbolizer({42: [(0x400000, 0x100000, 0x400000, str(exe))]})
    # non-PIE: file_addr == ip directly
    sym, dso = s.resolve(42, ip)
    assert sym.startswith("spin")
    assert "app.c:" in sym, sym


def test_fold_stacks_synthetic(tmp_path):
    """flame.fold_stacks on synthetic -g callchain samples: root-first
    folding with counts (flamegraph.folded contract)."""
    import struct

    import numpy as np

    from sofa_amd.preprocess.flame import fold_stacks, write_folded
    from sofa_amd.preprocess.scs import MAX_FRAMES, SAMPLE_CS_DTYPE, ScsFile

    scs = ScsFile()
    n = 6
    cs = np.zeros(n, dtype=SAMPLE_CS_DTYPE)
    cs["type"] = 6
    cs["size"] = SAMPLE_CS_DTYPE.itemsize
    cs["pid"] = 42
    cs["tid"] = 42
    cs["n_frames"] = 2
    for i in range(n):
        # leaf-first frames: leaf=0x2000+ variation, root=0x1000
        cs["frames"][i][0] = 0x2000 + (0x10 if i >= 4 else 0)
        cs["frames"][i][1] = 0x1000
    scs.samples_cs = cs
    scs.comms = {42: "worker"}
    out = fold_stacks(scs)
    assert sum(out.values()) == n
    # two distinct stacks: 4 samples of one, 2 of the other; root first
    counts = sorted(out.values())
    assert counts == [2, 4]
    for key in out:
        parts = key.split(";")
        assert parts[0] == "worker"  # comm root
    path = write_folded(scs, str(tmp_path))
    assert path and os.path.isfile(path)
    lines = open(path).read().splitlines()
    assert len(lines) == 2 and all(l.rsplit(" ", 1)[1].isdigit() for l in lines)


def test_monitor_main_subprocess(tmp_path):
    """record.monitor_main as the bench uses it: writes poller files and
    exits on SIGTERM."""
    import signal
    import subprocess
    import sys
    import time

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "sofa_amd.record.monitor_main",
         "--logdir", str(tmp_path), "--rate", "20", "--no-gpu"],
        cwd=repo,
    )
    time.sleep(1.5)
    proc.send_signal(signal.SIGTERM)
    assert proc.wait(timeout=10) == 0
    mp = os.path.join(str(tmp_path), "mpstat.txt")
    assert os.path.isfile(mp) and os.path.getsize(mp) > 0
    assert os.path.isfile(os.path.join(str(tmp_path), "vmstat.txt"))


def test_dwarf_line_table_v4(tmp_path):
    """The pre-v5 directory/file-table parser branch (-gdwarf-4)."""
    import subprocess

    from sofa_amd.preprocess.dwarf_lines import LineTable
    from sofa_amd.preprocess.symbols import read_elf_symbols

    csrc = tmp_path / "v4.c"
    csrc.write_text("int f(int x){return x+1;}\nint main(void){return f(1)&0;}\n")
    exe = tmp_path / "v4"
    subprocess.run(["gcc", "-gdwarf-4", "-O0", "-no-pie", str(csrc), "-o", str(exe)],
                   check=True, capture_output=True)
    lt = LineTable(str(exe))
    assert lt.addrs, "no rows from a DWARF v4 line program"
    syms = {n: a for a, _, n in read_elf_symbols(str(exe))}
    hit = lt.lookup(syms["f"] + 4)
    assert hit is not None and hit[0].endswith("v4.c") and hit[1] in (1, 2)


def test_dwarf_line_table_v2(tmp_path):
    import subprocess

    from sofa_amd.preprocess.dwarf_lines import LineTable

    csrc = tmp_path / "v2.c"
    csrc.write_text("int main(void){return 0;}\n")
    exe = tmp_path / "v2"
    r = subprocess.run(["gcc", "-gdwarf-2", "-O0", "-no-pie", str(csrc), "-o", str(exe)],
                       capture_output=True)
    if r.returncode != 0:
        import pytest as _pytest

        _pytest.skip("toolchain rejects -gdwarf-2")
    lt = LineTable(str(exe))
    assert lt.addrs
