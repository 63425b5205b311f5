"""Alias of reference types.py — same import path."""
from min_tfs_client_amd.types import DataType  # noqa: F401
