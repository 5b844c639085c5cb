"""API-compat shim matching the reference `cgx_utils` package."""
from .allreduce_hooks import CGXState, cgx_hook  # noqa: F401
