from byzpy_amd.engine.node.base import ByzantineNode, HonestNode, Node
from byzpy_amd.engine.node.actors import ByzantineNodeActor, HonestNodeActor, NodeActor

__all__ = [
    "Node",
    "HonestNode",
    "ByzantineNode",
    "NodeActor",
    "HonestNodeActor",
    "ByzantineNodeActor",
]
