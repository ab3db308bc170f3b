"""RL subsystem tests: env reference, GAE, DQN/PPO learning (CPU parts)."""

import numpy as np
import pytest
import torch

from ai_crypto_trader_amd.data.synthetic import candles_chl_v, generate_ohlcv
from ai_crypto_trader_amd.models.rl import (
    DQNAgent, N_OBS, PPOAgent, TradingVecEnvCPU, gae_reference,
)


@pytest.fixture(scope="module")
def market():
    return candles_chl_v(generate_ohlcv(4000, 2, seed=13))


def test_env_cpu_basics(market):
    env = TradingVecEnvCPU(market, n_envs=4, ep_len=128, seed=1)
    obs = env.reset()
    assert obs.shape == (4, N_OBS)
    rng = np.random.default_rng(0)
    eq_rewards = np.zeros(4)
    for _ in range(200):
        obs, r, d = env.step(rng.integers(0, 3, 4))
        assert np.isfinite(obs).all() and np.isfinite(r).all()
        eq_rewards += r
    # log-return rewards of a feeful random policy should be small negative
    assert (np.abs(eq_rewards) < 1.0).all()


def test_env_cpu_hold_zero_reward(market):
    """Never trading -> equity constant -> all rewards exactly 0."""
    env = TradingVecEnvCPU(market, n_envs=2, ep_len=64, seed=3)
    env.reset()
    for _ in range(100):
        _, r, _ = env.step(np.zeros(2, dtype=np.int64))
        np.testing.assert_array_equal(r, 0.0)


def test_env_cpu_roundtrip_fee(market):
    """Buy then immediately sell pays ~2x fee in log-return terms."""
    env = TradingVecEnvCPU(market, n_envs=1, ep_len=512, seed=4)
    env.reset()
    _, r_buy, _ = env.step(np.array([1]))
    _, r_sell, _ = env.step(np.array([2]))
    sym, t = int(env.state[0][0]), int(env.state[0][1])
    # combined cost ≈ 2 * fee + price move between the two candles
    move = np.log(env.candles[sym, t, 0] / env.candles[sym, t - 1, 0])
    assert abs((r_buy + r_sell) - (np.log((1 - 0.001) ** 2) + move)) < 1e-3


def test_gae_reference_analytic():
    """gamma=lam=1, no dones: advantage = sum of future deltas."""
    T, E = 5, 3
    torch.manual_seed(0)
    rew = torch.randn(T, E)
    val = torch.randn(T + 1, E)
    dones = torch.zeros(T, E)
    adv, ret = gae_reference(rew, val, dones, gamma=1.0, lam=1.0)
    expect = rew.flip(0).cumsum(0).flip(0) + val[-1] - val[:-1]
    torch.testing.assert_close(adv, expect, rtol=1e-5, atol=1e-5)
    # dones cut the recursion
    dones2 = torch.ones(T, E)
    adv2, _ = gae_reference(rew, val, dones2, gamma=1.0, lam=1.0)
    torch.testing.assert_close(adv2, rew - val[:-1], rtol=1e-5, atol=1e-5)


class _BanditEnv:
    """Tiny CPU env: action 1 always yields +1, others 0. For learning
    smoke tests without a GPU."""

    def __init__(self, n_envs=16, n_obs=N_OBS):
        self.n_envs = n_envs
        self.obs = torch.zeros(n_envs, n_obs)

    def reset(self):
        return self.obs

    def step(self, actions):
        r = (actions == 1).float()
        done = torch.ones(self.n_envs)
        return self.obs, r, done


def test_dqn_learns_bandit():
    torch.manual_seed(0)
    agent = DQNAgent("cpu", seed=0)
    env = _BanditEnv()
    obs = env.reset()
    for i in range(400):
        a = agent.act(obs)
        nxt, r, d = env.step(a)
        agent.remember(obs, a, r, nxt, d)
        agent.replay()
    with torch.no_grad():
        q = agent.q(obs[:1])
    assert int(q.argmax()) == 1, q


def test_ppo_learns_bandit():
    torch.manual_seed(0)
    agent = PPOAgent("cpu", seed=0, lr=1e-2, ent_coef=0.0)
    env = _BanditEnv()
    obs = env.reset()
    for _ in range(30):
        T, E = 16, env.n_envs
        obs_b = torch.zeros(T, E, N_OBS)
        act_b = torch.zeros(T, E, dtype=torch.long)
        logp_b = torch.zeros(T, E)
        rew_b = torch.zeros(T, E)
        done_b = torch.zeros(T, E)
        val_b = torch.zeros(T + 1, E)
        for t in range(T):
            a, logp, v = agent.policy(obs)
            obs_b[t], act_b[t], logp_b[t], val_b[t] = obs, a, logp, v
            _, r, d = env.step(a)
            rew_b[t], done_b[t] = r, d
        agent.update(obs_b, act_b, logp_b, rew_b, done_b, val_b)
    with torch.no_grad():
        logits, _ = agent.net(obs[:1])
    assert int(logits.argmax()) == 1, logits
