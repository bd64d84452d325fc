"""Search-space DSL (reference tune.choice at cell-57 NLP,
tune.uniform/randint at Introduction_to_Ray_AI_Runtime.ipynb:cell-45)."""
from __future__ import annotations

import math
import random
from dataclasses import dataclass


class Domain:
    def sample(self, rng: random.Random):
        raise NotImplementedError


@dataclass
class Choice(Domain):
    categories: list

    def sample(self, rng):
        return rng.choice(self.categories)


@dataclass
class Uniform(Domain):
    lower: float
    upper: float

    def sample(self, rng):
        return rng.uniform(self.lower, self.upper)


@dataclass
class QUniform(Domain):
    lower: float
    upper: float
    q: float

    def sample(self, rng):
        v = rng.uniform(self.lower, self.upper)
        return round(v / self.q) * self.q


@dataclass
class LogUniform(Domain):
    lower: float
    upper: float

    def sample(self, rng):
        return math.exp(rng.uniform(math.log(self.lower), math.log(self.upper)))


@dataclass
class RandInt(Domain):
    lower: int
    upper: int

    def sample(self, rng):
        return rng.randrange(self.lower, self.upper)


@dataclass
class QRandInt(Domain):
    lower: int
    upper: int
    q: int

    def sample(self, rng):
        return (rng.randrange(self.lower, self.upper) // self.q) * self.q


@dataclass
class RandN(Domain):
    mean: float = 0.0
    sd: float = 1.0

    def sample(self, rng):
        return rng.gauss(self.mean, self.sd)


@dataclass
class SampleFrom(Domain):
    fn: object

    def sample(self, rng):
        return self.fn(None)


@dataclass
class GridSearch:
    values: list


def choice(categories):
    return Choice(list(categories))


def uniform(lower, upper):
    return Uniform(lower, upper)


def quniform(lower, upper, q):
    return QUniform(lower, upper, q)


def loguniform(lower, upper):
    return LogUniform(lower, upper)


def randint(lower, upper):
    return RandInt(lower, upper)


def qrandint(lower, upper, q):
    return QRandInt(lower, upper, q)


def randn(mean=0.0, sd=1.0):
    return RandN(mean, sd)


def sample_from(fn):
    return SampleFrom(fn)


def grid_search(values):
    return GridSearch(list(values))


def sample_space(space, rng: random.Random):
    """Recursively sample one config from a param_space dict."""
    if isinstance(space, Domain):
        return space.sample(rng)
    if isinstance(space, GridSearch):
        return None  # placeholder — filled by the expand_grid overrides
    if isinstance(space, dict):
        return {k: sample_space(v, rng) for k, v in space.items()}
    if isinstance(space, list):
        return [sample_space(v, rng) for v in space]
    return space


def expand_grid(space):
    """Return list of partial configs for every grid_search combination."""
    grids = []

    def find(path, node):
        if isinstance(node, GridSearch):
            grids.append((path, node.values))
        elif isinstance(node, dict):
            for k, v in node.items():
                find(path + (k,), v)

    find((), space)
    if not grids:
        return [dict()]
    import itertools

    combos = []
    for values in itertools.product(*[v for _, v in grids]):
        combos.append({path: val for (path, _), val in zip(grids, values)})
    return combos


def apply_overrides(config: dict, overrides: dict):
    for path, val in overrides.items():
        node = config
        for key in path[:-1]:
            node = node.setdefault(key, {})
        node[path[-1]] = val
    return config
