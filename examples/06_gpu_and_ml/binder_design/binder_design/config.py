"""Design-run configuration (the dataclass config plane)."""
from dataclasses import dataclass


@dataclass
class DesignConfig:
    target: str = "MKTAYIAKQRQISFVKSHFSRQLEERLGLIEVQ"
    binder_len: int = 24
    population: int = 32
    generations: int = 6
    mutation_rate: float = 0.15
    seed: int = 0
