"""Shim: the reference layout's ai/models/graphsage.py -> nerrf_amd."""
from nerrf_amd.models.graphsage import GraphSAGET, SageConfig, SageLayer  # noqa: F401
