from mi355.runtime.actor import ActorClass, ActorHandle, ActorMethod

__all__ = ["ActorClass", "ActorHandle", "ActorMethod"]
