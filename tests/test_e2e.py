"""End-to-end CLI tests (CPU-only host; mirrors the reference's e2e matrix
test/test.py:62-78: run the real CLI, assert the Complete!! sentinel)."""

import os
import subprocess
import sys

import pytest

SOFA = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "bin", "sofa")


def run_sofa(args, timeout=120):
    return subprocess.run(
        [sys.executable, SOFA] + args,
        capture_output=True,
        text=True,
        timeout=timeout,
    )


def test_stat_dd_complete(tmp_path, native_built):
    """BASELINE config 1: sofa stat "dd ..." CPU-only plumbing."""
    logdir = str(tmp_path / "sofalog")
    out_file = str(tmp_path / "dummy.out")
    r = run_sofa(
        [
            "stat",
            f"dd if=/dev/zero of={out_file} bs=10M count=20",
            "--logdir",
            logdir,
            "--no_gpu",
        ]
    )
    assert r.returncode == 0, r.stderr
    assert "Complete!!" in r.stdout
    for artifact in [
        "cputrace.csv",
        "mpstat.csv",
        "report.js",
        "features.csv",
        "performance.csv",
        "misc.txt",
    ]:
        assert os.path.isfile(os.path.join(logdir, artifact)), artifact
    # sofaboard copied into logdir
    assert os.path.isfile(os.path.join(logdir, "index.html"))


def test_report_rerunnable(tmp_path, native_built):
    """Stages are idempotent over the logdir (reference --skip_preprocess)."""
    logdir = str(tmp_path / "sofalog")
    r = run_sofa(["stat", "sleep 1", "--logdir", logdir, "--no_gpu"])
    assert "Complete!!" in r.stdout, r.stderr
    r2 = run_sofa(["report", "--logdir", logdir, "--no_gpu"])
    assert "Complete!!" in r2.stdout, r2.stderr
    r3 = run_sofa(["analyze", "--logdir", logdir, "--no_gpu", "--skip_preprocess"])
    assert "Complete!!" in r3.stdout, r3.stderr


def test_clean(tmp_path, native_built):
    logdir = str(tmp_path / "sofalog")
    r = run_sofa(["stat", "sleep 0.5", "--logdir", logdir, "--no_gpu"])
    assert "Complete!!" in r.stdout
    r = run_sofa(["clean", "--logdir", logdir])
    assert r.returncode == 0
    assert not os.path.isfile(os.path.join(logdir, "cputrace.csv"))
    assert not os.path.isfile(os.path.join(logdir, "report.js"))


def test_record_needs_command():
    r = run_sofa(["record"])
    assert r.returncode == 2


def test_pystacks_e2e(tmp_path, native_built):
    logdir = str(tmp_path / "log")
    code = "import time\nt=time.time()\nx=0\nwhile time.time()-t<1.5: x+=1\n"
    script = tmp_path / "busy.py"
    script.write_text(code)
    r = run_sofa(
        ["stat", f"{sys.executable} {script}", "--logdir", logdir, "--no_gpu", "--enable_py_stacks"]
    )
    assert "Complete!!" in r.stdout, r.stderr
    import glob

    assert glob.glob(os.path.join(logdir, "pystacks.txt.*")), "no pystacks raw output"
    assert os.path.isfile(os.path.join(logdir, "pystacks.csv"))


def test_cluster_report_cli(tmp_path, native_built):
    """`sofa report --cluster_ip a,b` over two recorded per-node logdirs."""
    base = str(tmp_path / "clog")
    for ip in ("10.0.0.1", "10.0.0.2"):
        r = run_sofa(["stat", "sleep 0.6", "--logdir", f"{base}-{ip}", "--no_gpu"])
        assert "Complete!!" in r.stdout
    r = run_sofa(["report", "--logdir", base, "--cluster_ip", "10.0.0.1,10.0.0.2", "--no_gpu"])
    assert r.returncode == 0, r.stderr
    assert "Complete!!" in r.stdout
    assert os.path.isfile(os.path.join(base, "cluster_report.csv"))


def test_attach_mode(tmp_path, native_built):
    """`sofa record --attach PID` observes a running process."""
    import subprocess as sp

    logdir = str(tmp_path / "log")
    busy = sp.Popen([sys.executable, "-c",
                     "import time\nt=time.time()\nx=0\nwhile time.time()-t<3: x+=1"])
    try:
        r = run_sofa(["stat", "--attach", str(busy.pid), "--duration", "1.5",
                      "--logdir", logdir, "--no_gpu"])
        assert "Complete!!" in r.stdout, (r.stdout[-1500:], r.stderr[-800:])
        assert os.path.isfile(os.path.join(logdir, "cputrace.csv"))
    finally:
        busy.wait()


def test_demo_logdir_analyzes(tmp_path):
    """The committed demo logdir stays analyzable (guards the demo artifact);
    runs on a COPY so the committed files are never mutated."""
    import shutil

    demo = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "demo", "logdir")
    if not os.path.isdir(demo):
        import pytest

        pytest.skip("demo not present")
    work = str(tmp_path / "logdir")
    shutil.copytree(demo, work)
    r = run_sofa(["analyze", "--logdir", work, "--skip_preprocess", "--no_gpu"])
    assert "Complete!!" in r.stdout


def test_config_file_and_plugin(tmp_path, native_built):
    """YAML --config + --plugins load path through the real CLI."""
    cfgfile = tmp_path / "c.yaml"
    cfgfile.write_text("sys_mon_rate: 25\nnum_swarms: 4\n")
    plugdir = tmp_path / "plugs"
    plugdir.mkdir()
    (plugdir / "myplug.py").write_text(
        "def myplug(cfg):\n    print('PLUGIN-SAW', cfg.num_swarms)\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = str(plugdir) + ":" + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, SOFA, "stat", "sleep 0.4", "--logdir", str(tmp_path / "log"),
         "--no_gpu", "--config", str(cfgfile), "--plugins", "myplug"],
        capture_output=True, text=True, timeout=120, env=env,
    )
    assert "Complete!!" in r.stdout, r.stderr
    # plugins run AFTER --config is applied, so they see the YAML overrides
    assert "PLUGIN-SAW 4" in r.stdout


def test_duration_timebox(tmp_path, native_built):
    """--duration stops a long-running launched target."""
    import time as _time

    logdir = str(tmp_path / "log")
    t0 = _time.time()
    r = run_sofa(["stat", "sleep 30", "--logdir", logdir, "--no_gpu", "--duration", "1.5"])
    took = _time.time() - t0
    assert "Complete!!" in r.stdout, r.stderr
    assert took < 25, f"duration not enforced ({took:.1f}s)"


def test_diff_verb_end_to_end(tmp_path):
    """`sofa diff` through the CLI: two recorded-with-swarms logdirs ->
    swarm_diff.csv (reference bin/sofa:338-350 verb path)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sofa = os.path.join(repo, "bin", "sofa")
    logs = []
    for i, count in enumerate((6, 6)):
        logdir = str(tmp_path / f"log{i}")
        r = subprocess.run(
            [sys.executable, sofa, "stat",
             f"dd if=/dev/zero of={tmp_path}/d{i}.out bs=8M count={count}",
             "--logdir", logdir, "--enable_swarms", "--no_gpu"],
            capture_output=True, text=True, timeout=300, cwd=repo,
        )
        assert "Complete!!" in r.stdout, (r.stdout[-2000:], r.stderr[-1500:])
        logs.append(logdir)
    r = subprocess.run(
        [sys.executable, sofa, "diff", "--base_logdir", logs[0],
         "--match_logdir", logs[1], "--logdir", logs[1], "--skip_preprocess"],
        capture_output=True, text=True, timeout=300, cwd=repo,
    )
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-1500:])
    assert os.path.isfile(os.path.join(logs[1], "swarm_diff.csv")), r.stdout[-1500:]
