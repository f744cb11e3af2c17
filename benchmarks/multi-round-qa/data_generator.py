#!/usr/bin/env python3
"""Synthetic conversation dataset generator (ShareGPT-preprocessing
equivalent for the offline environment): emits a JSONL of multi-round
conversations shaped like the canonical workload."""
import argparse
import json
import random

WORDS = "alpha bravo charlie delta echo foxtrot golf hotel india".split()


def gen_text(n, rng):
    return " ".join(rng.choice(WORDS) for _ in range(n))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--num-users", type=int, default=320)
    ap.add_argument("--num-rounds", type=int, default=10)
    ap.add_argument("--system-tokens", type=int, default=1000)
    ap.add_argument("--history-tokens", type=int, default=20000)
    ap.add_argument("--question-tokens", type=int, default=100)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--output", default="conversations.jsonl")
    a = ap.parse_args()
    rng = random.Random(a.seed)
    system = gen_text(a.system_tokens, rng)
    with open(a.output, "w") as f:
        for u in range(a.num_users):
            conv = {
                "id": f"user-{u}",
                "system": system,
                "history": gen_text(a.history_tokens, rng),
                "questions": [
                    gen_text(a.question_tokens, rng)
                    for _ in range(a.num_rounds)
                ],
            }
            f.write(json.dumps(conv) + "\n")
    print(f"wrote {a.num_users} conversations to {a.output}")


if __name__ == "__main__":
    main()
