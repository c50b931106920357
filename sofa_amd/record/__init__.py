from .recorder import sofa_record, sofa_clean  # noqa: F401
