from byzpy_amd.configs.actor import get_actor, set_actor

__all__ = ["set_actor", "get_actor"]
