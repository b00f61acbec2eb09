from byzpy_amd.engine.peer_to_peer.topology import Topology
from byzpy_amd.engine.peer_to_peer.train import PeerToPeer

__all__ = ["Topology", "PeerToPeer"]
