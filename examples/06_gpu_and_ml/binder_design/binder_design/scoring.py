"""Binder scoring: a physics-flavored surrogate (hydropathy complementarity
+ charge pairing against the target) standing in for the folding oracle."""
KD = {"A": 1.8, "C": 2.5, "D": -3.5, "E": -3.5, "F": 2.8, "G": -0.4,
      "H": -3.2, "I": 4.5, "K": -3.9, "L": 3.8, "M": 1.9, "N": -3.5,
      "P": -1.6, "Q": -3.5, "R": -4.5, "S": -0.8, "T": -0.7, "V": 4.2,
      "W": -0.9, "Y": -1.3}
CHARGE = {"D": -1.0, "E": -1.0, "K": 1.0, "R": 1.0, "H": 0.5}


def score_binder(binder: str, target: str) -> float:
    """Higher is better: hydrophobic patches of the binder should face the
    target's, opposite charges should pair."""
    s = 0.0
    for i, b in enumerate(binder):
        t = target[i % len(target)]
        s += 0.1 * KD[b] * KD[t]
        s -= 0.5 * CHARGE.get(b, 0.0) * CHARGE.get(t, 0.0)
    return s / len(binder)
