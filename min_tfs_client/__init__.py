"""Drop-in alias for the reference package name.

``from min_tfs_client.requests import TensorServingClient`` works exactly as
with zendesk/min-tfs-client (reference tensor_serving_client/min_tfs_client/),
backed by the MI355X-native implementation in ``min_tfs_client_amd``.
"""
from min_tfs_client_amd import TensorServingClient  # noqa: F401
