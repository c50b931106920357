"""In-process Python stack sampler (pyflame replacement).

The reference prefixes the target with `pyflame --flamechart -o pystacks.txt`
(cyliustack/sofa bin/sofa_record.py:326-333); pyflame sampled the WHOLE
interpreter, so this does too: every thread in sys._current_frames() is
stacked per tick (round-1 verdict: main-thread-only sampling made dataloader
and worker threads invisible).  Injected via PYTHONPATH (sitecustomize
imports automatically in every Python child); a daemon thread samples at
SOFA_PYSTACKS_HZ (default 50).

Output format (one sample per two lines; header line grew tid + thread name
fields, the parser accepts both shapes):
    <epoch seconds> <tid> <thread-name>
    frameN;...;frame1;frame0
"""

import os

if os.environ.get("SOFA_PYSTACKS_OUT"):
    import sys
    import threading
    import time

    _out_path = os.environ["SOFA_PYSTACKS_OUT"]
    _hz = float(os.environ.get("SOFA_PYSTACKS_HZ", "50"))

    def _sampler():
        try:
            # line-buffered: the daemon thread dies with the process, so
            # anything still in a block buffer would be lost
            f = open(_out_path + ".%d" % os.getpid(), "w", buffering=1)
        except OSError:
            return
        period = 1.0 / max(_hz, 1.0)
        own_ident = threading.get_ident()
        names = {}
        names_refresh = 0.0
        while True:
            t = time.time()
            if t >= names_refresh:  # thread names change rarely; cache 1 s
                names = {th.ident: th.name for th in threading.enumerate()}
                names_refresh = t + 1.0
            for ident, frame in sys._current_frames().items():
                if ident == own_ident or frame is None:
                    continue
                if ident not in names:  # new thread since last refresh
                    names = {th.ident: th.name for th in threading.enumerate()}
                    names_refresh = t + 1.0
                stack = []
                fr = frame
                depth = 0
                while fr is not None and depth < 64:
                    code = fr.f_code
                    stack.append(
                        "%s (%s:%d)"
                        % (code.co_name, os.path.basename(code.co_filename), fr.f_lineno)
                    )
                    fr = fr.f_back
                    depth += 1
                if stack:
                    name = names.get(ident, "?").replace(" ", "_")
                    f.write(
                        "%.6f %d %s\n%s\n" % (t, ident, name, ";".join(reversed(stack)))
                    )
            dt = period - (time.time() - t)
            if dt > 0:
                time.sleep(dt)

    _thread = threading.Thread(target=_sampler, daemon=True, name="sofa-pystacks")
    _thread.start()
