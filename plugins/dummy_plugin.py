"""Example sofa plugin (reference plugins/dummy_plugin.py parity).

Load with:  sofa stat "<cmd>" --plugins dummy_plugin
(the plugins/ dir must be on PYTHONPATH; setup.py's editable install or
`export PYTHONPATH=$PWD/plugins` both work).  A plugin module exposes either
a function named like the module or `f(cfg)`; it receives the live SofaConfig
before the verb dispatch and may mutate it.
"""


def dummy_plugin(cfg):
    print(f"[dummy_plugin] hello — logdir is {cfg.logdir}")


f = dummy_plugin
