from .aisi import sofa_aisi  # noqa: F401
