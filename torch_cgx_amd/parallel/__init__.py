from . import partition, sra_sim  # noqa: F401
