"""Per-request block-IO tracing via tracefs (blktrace parity).

The reference shells out to blktrace/blkparse and matches D (dispatch) to C
(complete) rows for per-IO latency (cyliustack/sofa
bin/sofa_preprocess.py:684-781).  Rebuilt on the kernel's tracefs
block tracepoints directly — no blktrace binary, no per-device setup:

  * a private tracefs INSTANCE (instances/sofa_<pid>) so global tracing
    state is never touched;
  * trace_clock mono_raw — the same clock every other sofa stream stamps,
    so no extra clock pairing is needed;
  * events/block/block_rq_issue + block_rq_complete, streamed from
    trace_pipe into <logdir>/blktrace.txt.

Degrades silently when tracefs is unavailable (unprivileged container).
"""

from __future__ import annotations

import os
import threading
from typing import Optional

from .. import printing as p

TRACEFS_ROOTS = ("/sys/kernel/tracing", "/sys/kernel/debug/tracing")


def find_tracefs() -> Optional[str]:
    for root in TRACEFS_ROOTS:
        if os.path.isdir(root) and os.access(root, os.W_OK):
            return root
    return None


class BlkTracer(threading.Thread):
    """Streams block_rq_issue/complete tracepoints to logdir/blktrace.txt."""

    def __init__(self, logdir: str, device: str = ""):
        super().__init__(daemon=True, name="sofa-blkio")
        self.logdir = logdir
        self.device = device  # optional "8,0"-style or name filter (unused in v1)
        self.instance: Optional[str] = None
        self._stop_evt = threading.Event()
        self._pipe = None
        self.ok = False

    def _setup(self) -> bool:
        root = find_tracefs()
        if root is None:
            return False
        inst = os.path.join(root, "instances", f"sofa_{os.getpid()}")
        try:
            # reap instances from crashed/killed runs (rmdir only works when
            # the creator is gone; EBUSY for live ones is fine)
            inst_root = os.path.join(root, "instances")
            if os.path.isdir(inst_root):
                for name in os.listdir(inst_root):
                    if name.startswith("sofa_") and name != f"sofa_{os.getpid()}":
                        try:
                            pid = int(name.split("_", 1)[1])
                            if not os.path.exists(f"/proc/{pid}"):
                                os.rmdir(os.path.join(inst_root, name))
                        except (ValueError, OSError):
                            pass
            os.makedirs(inst, exist_ok=True)
            with open(os.path.join(inst, "trace_clock"), "w") as f:
                f.write("mono_raw")
            for ev in ("block_rq_issue", "block_rq_complete"):
                with open(
                    os.path.join(inst, "events", "block", ev, "enable"), "w"
                ) as f:
                    f.write("1")
            self.instance = inst
            return True
        except OSError as e:
            p.print_warning(f"blkio tracing unavailable: {e}")
            try:
                if os.path.isdir(inst):
                    os.rmdir(inst)
            except OSError:
                pass
            return False

    def run(self) -> None:
        if not self._setup():
            return
        self.ok = True
        out_path = os.path.join(self.logdir, "blktrace.txt")
        try:
            self._pipe = open(os.path.join(self.instance, "trace_pipe"), "rb", buffering=0)
            with open(out_path, "wb") as out:
                while not self._stop_evt.is_set():
                    data = self._pipe.read(65536)  # blocks until events arrive
                    if not data:
                        break
                    out.write(data)
        except OSError as e:
            p.print_warning(f"blkio stream ended: {e}")
        finally:
            self._teardown()

    def stop(self) -> None:
        self._stop_evt.set()
        # unblock the trace_pipe read: disabling events + closing from another
        # thread; a tiny sentinel write to the trace marker also wakes readers
        if self.instance:
            try:
                for ev in ("block_rq_issue", "block_rq_complete"):
                    with open(
                        os.path.join(self.instance, "events", "block", ev, "enable"),
                        "w",
                    ) as f:
                        f.write("0")
                with open(os.path.join(self.instance, "trace_marker"), "w") as f:
                    f.write("sofa-stop")
            except OSError:
                pass
        if self._pipe is not None:
            try:
                self._pipe.close()
            except OSError:
                pass
        self.join(timeout=2.0)

    def _teardown(self) -> None:
        if not self.instance:
            return
        try:
            os.rmdir(self.instance)
        except OSError:
            pass
        self.instance = None
