from .main import sofa_preprocess  # noqa: F401
