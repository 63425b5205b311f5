"""Wire layer: byte-compatible TF-Serving protobuf schema + message classes."""
from . import messages  # noqa: F401
from .messages import *  # noqa: F401,F403
