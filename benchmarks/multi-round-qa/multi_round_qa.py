#!/usr/bin/env python3
"""Multi-round QA serving benchmark (async, OpenAI-compatible endpoint).

Same workload shape and metric definitions as the reference harness
(reference benchmarks/multi-round-qa/multi-round-qa.py): N concurrent users
x M rounds of chat against a router URL, shared system prompt, growing
per-user history, Poisson-ish arrivals at a target QPS, streaming TTFT
capture, and a summary of QPS / input tokens/s / output tokens/s / TTFT
percentiles. Works against any OpenAI-compatible endpoint (ours or not).

Example:
  python multi_round_qa.py --base-url http://localhost:8001 --model m1 \
      --num-users 10 --num-rounds 5 --qps 0.5 --shared-system-prompt 1000 \
      --user-history-prompt 2000 --answer-len 100
"""

import argparse
import asyncio
import json
import random
import time
from dataclasses import dataclass, field
from typing import List, Optional

import aiohttp
import numpy as np

WORDS = (
    "alpha bravo charlie delta echo foxtrot golf hotel india juliet kilo "
    "lima mike november oscar papa quebec romeo sierra tango uniform victor "
    "whiskey xray yankee zulu"
).split()


def gen_text(n_tokens: int, rng: random.Random) -> str:
    return " ".join(rng.choice(WORDS) for _ in range(n_tokens))


@dataclass
class RequestRecord:
    launch_time: float = 0.0
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None
    prompt_tokens: int = 0
    generation_tokens: int = 0
    error: Optional[str] = None

    @property
    def ttft(self) -> Optional[float]:
        if self.first_token_time is None:
            return None
        return self.first_token_time - self.launch_time


@dataclass
class UserSession:
    user_id: int
    system_prompt: str
    history: List[dict] = field(default_factory=list)
    rounds_done: int = 0
    # real questions drawn from a ShareGPT conversation (else generated)
    questions: Optional[List[str]] = None


async def run_round(
    session: aiohttp.ClientSession,
    args,
    user: UserSession,
    records: List[RequestRecord],
    rng: random.Random,
) -> None:
    if user.questions and user.rounds_done < len(user.questions):
        question = user.questions[user.rounds_done]
    else:
        question = gen_text(args.question_len, rng)
    user.history.append({"role": "user", "content": question})
    messages = (
        [{"role": "system", "content": user.system_prompt}] + user.history
    )
    rec = RequestRecord(launch_time=time.time())
    records.append(rec)
    body = {
        "model": args.model,
        "messages": messages,
        "max_tokens": args.answer_len,
        "temperature": 0.0,
        "ignore_eos": True,
        "stream": True,
        "stream_options": {"include_usage": True},
    }
    answer = ""
    try:
        async with session.post(
            args.base_url.rstrip("/") + "/v1/chat/completions",
            json=body,
            headers={"x-user-id": f"user-{user.user_id}"},
            timeout=aiohttp.ClientTimeout(total=args.request_timeout),
        ) as resp:
            if resp.status != 200:
                rec.error = f"http {resp.status}"
                return
            buffer = ""
            async for chunk in resp.content.iter_any():
                if rec.first_token_time is None:
                    rec.first_token_time = time.time()
                buffer += chunk.decode(errors="replace")
                while "\n\n" in buffer:
                    event, buffer = buffer.split("\n\n", 1)
                    if not event.startswith("data: "):
                        continue
                    payload = event[6:]
                    if payload.strip() == "[DONE]":
                        continue
                    try:
                        data = json.loads(payload)
                    except json.JSONDecodeError:
                        continue
                    for choice in data.get("choices", []):
                        delta = choice.get("delta") or {}
                        answer += delta.get("content") or choice.get(
                            "text", ""
                        ) or ""
                    usage = data.get("usage")
                    if usage:
                        rec.prompt_tokens = usage.get("prompt_tokens", 0)
                        rec.generation_tokens = usage.get(
                            "completion_tokens", 0
                        )
    except (aiohttp.ClientError, asyncio.TimeoutError, OSError) as e:
        rec.error = f"{type(e).__name__}"
        return
    rec.finish_time = time.time()
    user.history.append({"role": "assistant", "content": answer})
    user.rounds_done += 1


async def user_task(args, user, session, records, rng, stop_time):
    # ramp-up stagger
    await asyncio.sleep(rng.random() * args.init_user_interval)
    while user.rounds_done < args.num_rounds and time.time() < stop_time:
        # Poisson arrival shaping across users: global qps / num users
        if args.qps > 0:
            await asyncio.sleep(
                rng.expovariate(args.qps / args.num_users)
            )
        await run_round(session, args, user, records, rng)
        # trim history to bound prompt growth
        max_msgs = 2 * args.max_rounds_kept
        if len(user.history) > max_msgs:
            user.history = user.history[-max_msgs:]


def summarize(records: List[RequestRecord], elapsed: float) -> dict:
    done = [r for r in records if r.finish_time is not None]
    ttfts = [r.ttft for r in done if r.ttft is not None]
    errors = [r for r in records if r.error]
    total_prompt = sum(r.prompt_tokens for r in done)
    total_gen = sum(r.generation_tokens for r in done)
    summary = {
        "finished_requests": len(done),
        "errors": len(errors),
        "elapsed_s": round(elapsed, 2),
        "processing_speed_rps": round(len(done) / elapsed, 3),
        "input_tokens_per_s": round(total_prompt / elapsed, 1),
        "output_tokens_per_s": round(total_gen / elapsed, 1),
        "ttft_avg_s": round(float(np.mean(ttfts)), 4) if ttfts else None,
        "ttft_p50_s": round(float(np.percentile(ttfts, 50)), 4)
        if ttfts
        else None,
        "ttft_p90_s": round(float(np.percentile(ttfts, 90)), 4)
        if ttfts
        else None,
        "ttft_p99_s": round(float(np.percentile(ttfts, 99)), 4)
        if ttfts
        else None,
    }
    return summary


def load_sharegpt(path: str, num_users: int, rng) -> List[List[str]]:
    """ShareGPT-format JSON -> per-user human-turn question lists
    (reference benchmarks/multi-round-qa/data_preprocessing.py role:
    drive the multi-round workload with real conversation turns)."""
    with open(path) as f:
        data = json.load(f)
    convs = []
    for item in data:
        turns = item.get("conversations") or item.get("items") or []
        qs = [t.get("value", "") for t in turns
              if t.get("from") in ("human", "user") and t.get("value")]
        if qs:
            convs.append(qs)
    if not convs:
        raise ValueError(f"no usable conversations in {path}")
    rng.shuffle(convs)
    return [convs[u % len(convs)] for u in range(num_users)]


async def main_async(args) -> dict:
    rng = random.Random(args.seed)
    system_prompt = gen_text(args.shared_system_prompt, rng)
    sharegpt = (
        load_sharegpt(args.sharegpt_file, args.num_users, rng)
        if getattr(args, "sharegpt_file", None) else None
    )
    users = []
    for u in range(args.num_users):
        user = UserSession(u, system_prompt)
        if sharegpt:
            user.questions = sharegpt[u]
        if args.user_history_prompt > 0:
            user.history.append(
                {
                    "role": "user",
                    "content": gen_text(args.user_history_prompt, rng),
                }
            )
            user.history.append({"role": "assistant", "content": "ok"})
        users.append(user)
    records: List[RequestRecord] = []
    t0 = time.time()
    stop_time = t0 + args.time_limit
    connector = aiohttp.TCPConnector(limit=0)
    async with aiohttp.ClientSession(connector=connector) as session:
        await asyncio.gather(
            *(
                user_task(args, u, session, records, rng, stop_time)
                for u in users
            )
        )
    elapsed = time.time() - t0
    return summarize(records, elapsed)


def parse_args(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--base-url", required=True)
    ap.add_argument("--model", required=True)
    ap.add_argument("--num-users", type=int, default=10)
    ap.add_argument("--num-rounds", type=int, default=5)
    ap.add_argument("--qps", type=float, default=0.5,
                    help="aggregate request arrival rate; 0 = closed loop")
    ap.add_argument("--shared-system-prompt", type=int, default=1000)
    ap.add_argument("--user-history-prompt", type=int, default=2000)
    ap.add_argument("--question-len", type=int, default=100)
    ap.add_argument("--answer-len", type=int, default=100)
    ap.add_argument("--time-limit", type=float, default=100.0)
    ap.add_argument("--init-user-interval", type=float, default=2.0)
    ap.add_argument("--max-rounds-kept", type=int, default=20)
    ap.add_argument("--request-timeout", type=float, default=600.0)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--sharegpt-file", default=None,
                    help="ShareGPT-format JSON: drive questions with "
                         "real conversation turns")
    ap.add_argument("--output", default=None, help="write summary JSON here")
    return ap.parse_args(argv)


def main():
    args = parse_args()
    summary = asyncio.run(main_async(args))
    print(json.dumps(summary, indent=2))
    if args.output:
        with open(args.output, "w") as f:
            json.dump(summary, f)


if __name__ == "__main__":
    main()
