from byzpy_amd.actor.base import ActorBackend, ActorRef
from byzpy_amd.actor.channels import ChannelRef, Endpoint, open_channel
from byzpy_amd.actor.factory import resolve_backend
from byzpy_amd.actor.router import channel_router

__all__ = [
    "ActorBackend",
    "ActorRef",
    "Endpoint",
    "ChannelRef",
    "open_channel",
    "resolve_backend",
    "channel_router",
]
