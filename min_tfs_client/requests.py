"""Alias of reference requests.py:22-110 — same import path, same API."""
from min_tfs_client_amd.client import TensorServingClient  # noqa: F401
