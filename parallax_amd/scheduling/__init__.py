"""Cluster-level scheduling brain: layer allocation, request routing, node
lifecycle. Pure Python with zero I/O — unit-testable with fabricated nodes
(reference analogue: src/scheduling/; same separation, fresh implementation)."""

from .model_info import ModelInfo
from .node import Node, NodeHardware
from .scheduler import ClusterScheduler
