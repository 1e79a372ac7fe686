# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/distributed_mcts.py"]
# ---
# # Distributed Monte-Carlo tree search (misc/distributed_mcts_reasoner.py role)
#
# The search tree lives in the client; rollouts fan out with `.map` (one
# batch of leaves per wave).  The toy domain: choose digits to approach a
# target number — any expensive simulator slots into `rollout`.

import math
import random

import modal_examples_amd as modal

app = modal.App("example-mcts")

TARGET = 0.5612


@app.function()
def rollout(prefix: list) -> float:
    """Play out a random completion; reward = closeness to TARGET."""
    rng = random.Random(hash(tuple(prefix)) & 0xFFFF)
    digits = list(prefix) + [rng.randint(0, 9) for _ in range(6 - len(prefix))]
    value = sum(d * 10 ** -(i + 1) for i, d in enumerate(digits))
    return -abs(value - TARGET)


class Node:
    def __init__(self, prefix):
        self.prefix = prefix
        self.children = {}
        self.n = 0
        self.w = 0.0

    def ucb(self, parent_n):
        if self.n == 0:
            return float("inf")
        return self.w / self.n + 0.3 * math.sqrt(math.log(parent_n + 1) / self.n)


@app.local_entrypoint()
def main(iterations: int = 6, batch: int = 8):
    root = Node([])
    for it in range(iterations):
        # selection: collect `batch` leaves by UCB descent
        leaves = []
        for _ in range(batch):
            node = root
            while node.children and len(node.prefix) < 4:
                node = max(node.children.values(),
                           key=lambda c: c.ucb(node.n))
            if len(node.prefix) < 4 and node.n > 0 and not node.children:
                for d in range(0, 10, 3):
                    node.children[d] = Node(node.prefix + [d])
                node = list(node.children.values())[0]
            leaves.append(node)
        # parallel rollouts across the pool
        rewards = list(rollout.map([leaf.prefix for leaf in leaves]))
        for leaf, r in zip(leaves, rewards):
            n = leaf
            while True:
                n.n += 1
                n.w += r
                parent = _parent_of(root, n)
                if parent is None:
                    break
                n = parent
        best = max((c for c in root.children.values()), default=None,
                   key=lambda c: c.n)
        if best:
            print(f"iter {it}: best first digit {best.prefix} "
                  f"visits={best.n} value={best.w / max(1, best.n):.4f}")


def _parent_of(root, node, cur=None):
    cur = cur or root
    for c in cur.children.values():
        if c is node:
            return cur
        deep = _parent_of(root, node, c)
        if deep is not None:
            return deep
    return None
