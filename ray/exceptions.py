"""ray.exceptions compat (reference: ray.exceptions.RayActorError /
RayTaskError raised from ray.get on failed tasks/actors)."""
from mi355.runtime.actor import ActorDiedError

RayActorError = ActorDiedError


class RayTaskError(RuntimeError):
    pass
