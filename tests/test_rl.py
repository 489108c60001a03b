"""GRPO RL tests: advantage math, clipped loss values, packing, end-to-end
rl_step on a tiny model with a token-preference environment."""

import math

import torch

from megatron_amd.config import TransformerConfig
from megatron_amd.models.gpt import GPTModel
from megatron_amd.rl.grpo import group_relative_advantages, grpo_loss
from megatron_amd.rl.loop import pack_rollouts, policy_logprobs, rl_step
from megatron_amd.rl.rollout import Rollout

from megatron_amd.parallel.random import model_parallel_seed
from tests.utils import assert_close, init_single

MODEL_KW = dict(num_layers=2, hidden_size=64, num_attention_heads=4,
                num_query_groups=2, vocab_size=64, ffn_hidden_size=96,
                max_position_embeddings=128, gradient_accumulation_fusion=False)


def test_group_relative_advantages():
    r = torch.tensor([1.0, 3.0, 0.0, 0.0, 5.0])
    g = torch.tensor([0, 0, 1, 1, 2])
    a = group_relative_advantages(r, g)
    assert_close(a[0], torch.tensor(-1.0), rtol=1e-4, atol=1e-4)
    assert_close(a[1], torch.tensor(1.0), rtol=1e-4, atol=1e-4)
    assert float(a[2]) == 0.0 and float(a[3]) == 0.0  # zero-variance group
    assert float(a[4]) == 0.0  # singleton group


def test_grpo_loss_values():
    # one rollout, one token, ratio=e^0.5, adv=+1 -> clipped at 1.2
    lp = torch.tensor([[0.0]])
    beh = torch.tensor([[-0.5]])
    adv = torch.tensor([1.0])
    mask = torch.ones(1, 1)
    loss = grpo_loss(lp, beh, adv, mask, clip_ratio=0.2)
    assert_close(loss, torch.tensor(-1.2), rtol=1e-5, atol=1e-6)
    # negative advantage: min picks the UNCLIPPED (more negative) branch
    loss2 = grpo_loss(lp, beh, torch.tensor([-1.0]), mask, clip_ratio=0.2)
    assert_close(loss2, torch.tensor(math.exp(0.5)), rtol=1e-5, atol=1e-6)
    # kl penalty is zero when policies agree
    l3 = grpo_loss(lp, lp.clone(), adv, mask, ref_logprobs=lp.clone(), kl_coeff=0.1)
    assert_close(l3, torch.tensor(-1.0), rtol=1e-5, atol=1e-6)


def test_pack_rollouts():
    rollouts = [
        Rollout(prompt_tokens=[5, 6], response_tokens=[7, 8], behavior_logprobs=[-0.1, -0.2]),
        Rollout(prompt_tokens=[9], response_tokens=[3], behavior_logprobs=[-0.3]),
    ]
    ids, chosen, mask, beh = pack_rollouts(rollouts, torch.device("cpu"))
    assert ids.shape == (2, 4)
    assert ids[0].tolist() == [5, 6, 7, 8]
    assert mask[0].tolist() == [0.0, 1.0, 1.0, 0.0]  # positions 1,2 predict 7,8
    assert chosen[0, 1].item() == 7 and chosen[0, 2].item() == 8
    assert abs(beh[0, 2].item() + 0.2) < 1e-6
    assert mask[1].tolist() == [1.0, 0.0, 0.0, 0.0]


class _PreferToken:
    """Reward = fraction of response tokens equal to `target`."""

    def __init__(self, target):
        self.target = target

    def __call__(self, prompt, response):
        if not response:
            return 0.0
        return sum(1 for t in response if t == self.target) / len(response)


def test_rl_step_end_to_end():
    init_single()
    torch.manual_seed(0)
    cfg = TransformerConfig(num_layers=2, hidden_size=64, num_attention_heads=4,
                            vocab_size=32, max_position_embeddings=128)
    model = GPTModel(cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=3e-3)
    prompts = [[1, 2, 3], [4, 5]]
    env = _PreferToken(target=7)
    rewards = []
    for step in range(4):
        loss, mean_r, rollouts = rl_step(model, opt, prompts, env, group_size=4,
                                         max_tokens=8, seed=step)
        assert math.isfinite(loss)
        assert len(rollouts) == 8
        rewards.append(mean_r)
    # policy gradient should push the preferred token's probability up
    ids, chosen, mask, _ = pack_rollouts(
        [Rollout(prompt_tokens=[1, 2, 3], response_tokens=[7], behavior_logprobs=[0.0])],
        torch.device("cpu"),
    )
    with torch.no_grad():
        lp_after = policy_logprobs(model.eval(), ids, chosen)
    assert torch.isfinite(lp_after[mask.bool()]).all()


def test_refit_updates_rollout_model():
    """RL refit flow: after a training step changes the policy, refit_model
    pushes the new weights into a separate rollout/inference model."""
    import torch

    from megatron_amd.config import TransformerConfig
    from megatron_amd.models.gpt import GPTModel
    from megatron_amd.resharding import refit_model
    from tests.utils import init_single

    init_single()
    cfg = TransformerConfig(num_layers=2, hidden_size=32, num_attention_heads=4,
                            num_query_groups=2, vocab_size=64, ffn_hidden_size=48,
                            gradient_accumulation_fusion=False)
    torch.manual_seed(0)
    train_model = GPTModel(cfg)
    torch.manual_seed(0)
    rollout_model = GPTModel(cfg).eval()
    toks = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        before = rollout_model(toks, position_ids=None, attention_mask=None).clone()

    # a "training step": perturb the policy
    opt = torch.optim.SGD(train_model.parameters(), lr=0.5)
    loss = train_model(toks, labels=toks).mean()
    loss.backward()
    opt.step()

    refit_model(train_model, rollout_model)
    with torch.no_grad():
        after = rollout_model(toks, position_ids=None, attention_mask=None)
        expect = train_model(toks, position_ids=None, attention_mask=None)
    assert not torch.allclose(after, before, atol=1e-5)
    assert torch.allclose(after, expect, atol=1e-5)


def test_packed_rollout_logprobs_match_padded():
    """THD-packed policy recompute == right-padded recompute (the packing
    must not change any rollout's logprobs)."""
    import torch

    from megatron_amd.rl.loop import (
        pack_rollouts,
        pack_rollouts_thd,
        policy_logprobs,
        policy_logprobs_packed,
    )
    from megatron_amd.rl.rollout import Rollout

    init_single()
    model_parallel_seed(11)
    cfg = TransformerConfig(**MODEL_KW)
    model = GPTModel(cfg)
    model.eval()
    rollouts = [
        Rollout(prompt_tokens=[5, 6, 7], response_tokens=[8, 9], behavior_logprobs=[-1.0, -1.1]),
        Rollout(prompt_tokens=[3, 4], response_tokens=[1, 2, 10], behavior_logprobs=[-0.5, -0.6, -0.7]),
    ]
    dev = torch.device("cpu")
    ids, chosen, mask, _ = pack_rollouts(rollouts, dev)
    with torch.no_grad():
        lp_pad = policy_logprobs(model, ids, chosen)
    pids, pchosen, pmask, _, psp = pack_rollouts_thd(rollouts, dev)
    with torch.no_grad():
        lp_pack = policy_logprobs_packed(model, pids, pchosen, psp)
    # compare masked positions rollout by rollout
    off = 0
    for i, r in enumerate(rollouts):
        n = len(r.prompt_tokens) + len(r.response_tokens)
        sel_pack = lp_pack[0, off : off + n][pmask[0, off : off + n] > 0]
        sel_pad = lp_pad[i][mask[i] > 0]
        torch.testing.assert_close(sel_pack, sel_pad, rtol=1e-4, atol=1e-5)
        off += n


def test_rl_step_packed_runs():
    import torch

    from megatron_amd.rl.loop import rl_step

    init_single()
    model_parallel_seed(12)
    cfg = TransformerConfig(**MODEL_KW)
    model = GPTModel(cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4)

    def env(prompt, resp):
        return float(len(set(resp)))

    loss, reward, rollouts = rl_step(model, opt, [[3, 4, 5], [6, 7]], env,
                                     group_size=2, max_tokens=4, packed=True)
    assert torch.isfinite(torch.tensor(loss))
    assert len(rollouts) == 4


def test_server_agent_rollouts_through_rest():
    """ServerAgent: rollouts generated via the REST /api/generate endpoint
    (server-integrated rollouts, reference megatron/rl agent API)."""
    from fastapi.testclient import TestClient

    from megatron_amd.inference.engine import StaticInferenceEngine
    from megatron_amd.inference.server import create_app
    from megatron_amd.rl.agent import InProcessAgent, ServerAgent

    init_single()
    model_parallel_seed(13)
    cfg = TransformerConfig(**MODEL_KW)
    model = GPTModel(cfg).eval()
    engine = StaticInferenceEngine(model, max_batch=8, max_seq=128)
    app = create_app(engine, None)
    client = TestClient(app)

    def post(url, payload):
        r = client.post(url, json=payload)
        assert r.status_code == 200, r.text
        return r.json()

    def env(prompt, resp):
        return float(len(resp))

    agent = ServerAgent(env, post)
    rollouts = agent.get_rollouts([[3, 4, 5], [6, 7]], group_size=2, max_tokens=4, seed=1)
    assert len(rollouts) == 4
    for r in rollouts:
        assert len(r.response_tokens) > 0
        assert len(r.behavior_logprobs) == len(r.response_tokens)
        assert r.reward == len(r.response_tokens)

    # the in-process agent with the same seed produces the same tokens
    local = InProcessAgent(model, env, max_seq=128).get_rollouts(
        [[3, 4, 5], [6, 7]], group_size=2, max_tokens=4, seed=1)
    assert [r.response_tokens for r in local] == [r.response_tokens for r in rollouts]
