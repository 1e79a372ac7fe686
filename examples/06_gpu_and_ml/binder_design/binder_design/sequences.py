"""Sequence generation and mutation operators."""
import random

AA = "ACDEFGHIKLMNPQRSTVWY"


def random_binder(length: int, rng: random.Random) -> str:
    return "".join(rng.choice(AA) for _ in range(length))


def mutate(seq: str, rate: float, rng: random.Random) -> str:
    return "".join(rng.choice(AA) if rng.random() < rate else c for c in seq)
