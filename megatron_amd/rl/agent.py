"""Agent / environment abstraction for RL rollouts.

Capability analog of reference megatron/rl/agent/api.py: an Agent produces
scored rollouts for a batch of prompts; the trainer neither knows nor cares
whether generation ran in-process, against a serving endpoint, or came from
a replay source.

  * InProcessAgent  — the colocated path: StaticInferenceEngine on the
    training model (weights always current; refit-free).
  * ServerAgent     — server-integrated rollouts: generation via the REST
    /api/generate endpoint of a (possibly remote, disaggregated) serving
    deployment; weights reach the server via resharding/refit.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

from megatron_amd.rl.rollout import Environment, Rollout, generate_rollouts


class Agent:
    """Produces scored rollouts for a batch of prompts."""

    def get_rollouts(self, prompts: Sequence[List[int]], group_size: int = 4,
                     max_tokens: int = 32, temperature: float = 1.0,
                     seed: int = 0) -> List[Rollout]:
        raise NotImplementedError


class InProcessAgent(Agent):
    def __init__(self, model, env, max_seq: int = 2048):
        self.model = model
        self.env = env
        self.max_seq = max_seq

    def get_rollouts(self, prompts, group_size=4, max_tokens=32, temperature=1.0, seed=0):
        return generate_rollouts(self.model, prompts, self.env, group_size=group_size,
                                 max_tokens=max_tokens, temperature=temperature,
                                 seed=seed, max_seq=self.max_seq)


class ServerAgent(Agent):
    """Rollouts through a REST text-generation server.

    `post` is any callable (url, json) -> response-dict — an httpx/requests
    session against a remote server, or a FastAPI TestClient in tests."""

    def __init__(self, env, post: Callable[[str, dict], dict],
                 endpoint: str = "/api/generate"):
        self.env = env
        self.post = post
        self.endpoint = endpoint

    def get_rollouts(self, prompts, group_size=4, max_tokens=32, temperature=1.0, seed=0):
        score = self.env.score if isinstance(self.env, Environment) else self.env
        batch = [list(p) for p in prompts for _ in range(group_size)]
        resp = self.post(self.endpoint, {
            "prompts": batch, "max_tokens": max_tokens, "temperature": temperature,
            "logprobs": True, "seed": seed, "stop_on_eod": False,
        })
        rollouts = []
        for i, gen in enumerate(resp["generations"]):
            g = i // group_size
            prompt = batch[i]
            out_tokens = list(gen["tokens"])
            rollouts.append(Rollout(
                prompt_tokens=prompt,
                response_tokens=out_tokens,
                behavior_logprobs=list(gen.get("logprobs") or [0.0] * len(out_tokens)),
                reward=float(score(prompt, out_tokens)),
                group=g,
            ))
        return rollouts
