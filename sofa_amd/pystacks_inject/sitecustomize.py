"""In-process Python stack sampler (pyflame replacement).

The reference prefixes the target with `pyflame --flamechart -o pystacks.txt`
(cyliustack/sofa bin/sofa_record.py:326-333); pyflame is dead/absent, so
sofa_record injects this module via PYTHONPATH (sitecustomize imports
automatically in every Python child) and a daemon thread samples
sys._current_frames() at SOFA_PYSTACKS_HZ (default 50).

Output format (one sample per two lines, matching the reference's
pystacks.txt shape parsed at bin/sofa_preprocess.py:1709-1761):
    <epoch seconds>
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
        main_thread = threading.main_thread().ident
        while True:
            t = time.time()
            frames = sys._current_frames()
            frame = frames.get(main_thread)
            if frame is not None:
                stack = []
                fr = frame
                depth = 0
                while fr is not None and depth < 64:
                    code = fr.f_code
                    stack.append("%s (%s:%d)" % (code.co_name, os.path.basename(code.co_filename), fr.f_lineno))
                    fr = fr.f_back
                    depth += 1
                if stack and "_sampler" not in stack[0]:
                    f.write("%.6f\n%s\n" % (t, ";".join(reversed(stack))))
            dt = period - (time.time() - t)
            if dt > 0:
                time.sleep(dt)

    _thread = threading.Thread(target=_sampler, daemon=True, name="sofa-pystacks")
    _thread.start()
