"""Alias of reference constants.py — same import path."""
from min_tfs_client_amd.constants import (  # noqa: F401
    ENUM_TO_TF_MAPPING,
    NP_TO_ENUM_MAPPING,
    NP_TO_TF_MAPPING,
    NUMERICAL_TYPES,
    TF_TO_NP_MAPPING,
)
