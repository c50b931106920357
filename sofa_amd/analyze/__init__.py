from .main import sofa_analyze, cluster_analyze  # noqa: F401
