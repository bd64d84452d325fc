from mi355.runtime import ActorPool

__all__ = ["ActorPool"]
