"""RL agents: DQN (reference parity) + PPO on GPU-resident environments.

Replaces services/reinforcement_learning.py:
  :99-131   DQN Q-net 2x24 Dense + target net      -> DQNAgent (torch)
  :78, :335-419  replay deque(10000), batch-64 TD(0), target sync 100
  (BASELINE also asks for PPO)                     -> PPOAgent: 256
  synthetic-market envs/GPU stepped by the HIP env kernel
  (ops/hip/rl_env.hip), GAE(lambda) advantages via the hand-written
  reverse-scan kernel, DP gradient all-reduce over RCCL.

TradingVecEnv wraps the HIP env kernels on GPU; TradingVecEnvCPU is the
bit-compatible numpy reference used for golden tests (same Philox streams
for resets).
"""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

from ..ops import require_hip_ops
from ..parallel import dist as pdist

N_OBS = 12
N_ACT = 3
N_STATE = 16


class TradingVecEnv:
    """E GPU-resident trading environments over a shared candle tensor."""

    def __init__(self, candles: torch.Tensor, n_envs: int = 256,
                 ep_len: int = 1024, fee: float = 0.001, seed: int = 0):
        assert candles.is_cuda and candles.dtype == torch.float32
        self.ops = require_hip_ops()
        self.candles = candles.contiguous()
        self.nsym, self.T, _ = candles.shape
        self.n_envs = n_envs
        self.ep_len = ep_len
        self.fee = fee
        self.seed = seed
        self.epoch = 0
        dev = candles.device
        self.device = dev
        self.state = torch.zeros(n_envs, N_STATE, device=dev)
        self.obs = torch.zeros(n_envs, N_OBS, device=dev)
        self.reward = torch.zeros(n_envs, device=dev)
        self.done = torch.zeros(n_envs, device=dev)

    def _stream(self):
        return torch.cuda.current_stream(self.device).cuda_stream

    def reset(self) -> torch.Tensor:
        self.epoch += 1
        self.ops.env_reset(
            self.candles.data_ptr(), self.state.data_ptr(),
            self.obs.data_ptr(), self.nsym, self.T, self.n_envs,
            self.ep_len, self.seed, self.epoch, self._stream(),
        )
        return self.obs

    def step(self, actions: torch.Tensor):
        self.ops.env_step(
            self.candles.data_ptr(), self.state.data_ptr(),
            actions.to(torch.int32).contiguous().data_ptr(),
            self.obs.data_ptr(), self.reward.data_ptr(),
            self.done.data_ptr(), self.nsym, self.T, self.n_envs,
            self.ep_len, self.fee, self.seed, self.epoch, self._stream(),
        )
        return self.obs, self.reward, self.done


class TradingVecEnvCPU:
    """numpy reference of the HIP env kernel (same update order; resets are
    driven by the same Philox stream via ops.montecarlo.philox4x32_np)."""

    def __init__(self, candles: np.ndarray, n_envs: int = 8,
                 ep_len: int = 256, fee: float = 0.001, seed: int = 0):
        self.candles = np.asarray(candles, np.float32)
        self.nsym, self.T, _ = self.candles.shape
        self.n_envs, self.ep_len, self.fee = n_envs, ep_len, fee
        self.seed = seed
        self.epoch = 0
        self.state = np.zeros((n_envs, N_STATE), np.float32)
        self.obs = np.zeros((n_envs, N_OBS), np.float32)

    def _reset_env(self, e, rx, ry):
        st = self.state[e]
        sym = int(rx % np.uint32(self.nsym))
        max_start = max(self.T - self.ep_len - 32, 1)
        t0 = 16 + int(ry % np.uint32(max_start))
        c0 = self.candles[sym, t0, 0]
        st[:] = 0.0
        st[0], st[1] = sym, t0
        st[2] = st[3] = c0
        st[7] = c0
        st[8] = 1.0
        st[10] = c0
        st[11] = 1.0
        st[13] = 1.0
        st[12] = self.ep_len

    def reset(self):
        from ..ops.montecarlo import philox4x32_np

        self.epoch += 1
        e = np.arange(self.n_envs, dtype=np.uint64)
        c0, c1, _, _ = philox4x32_np(
            self.seed, e, np.full(self.n_envs, self.epoch, np.uint64)
        )
        for i in range(self.n_envs):
            self._reset_env(i, c0[i], c1[i])
            self.obs[i] = self._make_obs(i, 50.0, 0.0, 0.0, 0.0, 0.0,
                                         self.state[i][7])
        return self.obs.copy()

    def _make_obs(self, e, rsi, macd_hist, r1, r5, r15, close):
        st = self.state[e]
        in_pos = 1.0 if st[9] > 0 else 0.0
        upnl = (close / st[10] - 1.0) if in_pos else 0.0
        return np.array([
            rsi * 0.01, macd_hist / close * 100.0, r1 * 100.0, r5 * 100.0,
            r15 * 100.0, in_pos, upnl * 10.0, st[11] - 1.0,
            st[5] * 1000.0, st[6] * 1000.0, (st[2] / st[3] - 1.0) * 100.0,
            st[12] * 0.001,
        ], np.float32)

    def step(self, actions):
        from ..ops.montecarlo import philox4x32_np

        rewards = np.zeros(self.n_envs, np.float32)
        dones = np.zeros(self.n_envs, np.float32)
        f32 = np.float32
        for e in range(self.n_envs):
            st = self.state[e]
            sym, t = int(st[0]), int(st[1]) + 1
            close = f32(self.candles[sym, t, 0])
            a_f, a_s, a_sig = f32(2 / 13), f32(2 / 27), f32(2 / 10)
            st[2] = st[2] + a_f * (close - st[2])
            st[3] = st[3] + a_s * (close - st[3])
            macd = st[2] - st[3]
            st[4] = st[4] + a_sig * (macd - st[4])
            change = close - st[7]
            st[5] = st[5] + (max(change, 0) - st[5]) / f32(14)
            st[6] = st[6] + (max(-change, 0) - st[6]) / f32(14)
            rsi = 100.0 - 100.0 / (1.0 + st[5] / max(st[6], 1e-9))
            act = int(actions[e])
            if act == 1 and st[9] == 0.0:
                st[9] = st[8] * (1.0 - self.fee) / close
                st[10] = close
                st[8] = 0.0
            elif act == 2 and st[9] > 0.0:
                st[8] += st[9] * close * (1.0 - self.fee)
                st[9] = 0.0
            equity = st[8] + st[9] * close
            prev_eq = st[11]
            rewards[e] = np.log(max(equity, 1e-9) / max(prev_eq, 1e-9))
            st[12] -= 1.0
            ep_done = st[12] <= 0.0 or t + 2 >= self.T
            r1 = close / st[7] - 1.0
            c5 = self.candles[sym, max(t - 5, 0), 0]
            c15 = self.candles[sym, max(t - 15, 0), 0]
            st[7] = close
            st[11] = equity
            st[13] = prev_eq
            st[1] = t
            dones[e] = 1.0 if ep_done else 0.0
            self.obs[e] = self._make_obs(e, rsi, macd - st[4], r1,
                                         close / c5 - 1.0,
                                         close / c15 - 1.0, close)
            if ep_done:
                c0, c1, _, _ = philox4x32_np(
                    self.seed, np.array([e], np.uint64),
                    np.array([0x100000000 * self.epoch + t], np.uint64),
                )
                self._reset_env(e, c0[0], c1[0])
        return self.obs.copy(), rewards, dones


# ---------------------------------------------------------------------------


class QNet(nn.Module):
    """Reference-parity Q-network: Dense(24)-Dense(24)-Dense(n_act)
    (reinforcement_learning.py:113-124)."""

    def __init__(self, n_obs: int = N_OBS, n_act: int = N_ACT,
                 width: int = 24):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(n_obs, width), nn.ReLU(),
            nn.Linear(width, width), nn.ReLU(),
            nn.Linear(width, n_act),
        )

    def forward(self, x):
        return self.net(x)


class DQNAgent:
    """DQN with GPU-resident replay (reinforcement_learning.py semantics:
    replay 10k, batch 64, eps-greedy, target sync every 100 updates)."""

    def __init__(self, device, n_obs=N_OBS, n_act=N_ACT, lr=1e-3,
                 gamma=0.95, buffer_size=10_000, batch=64,
                 target_sync=100, seed=0):
        self.device = torch.device(device)
        torch.manual_seed(seed)
        self.q = QNet(n_obs, n_act).to(device)
        self.target = QNet(n_obs, n_act).to(device)
        self.target.load_state_dict(self.q.state_dict())
        self.opt = torch.optim.Adam(self.q.parameters(), lr=lr)
        self.gamma = gamma
        self.batch = batch
        self.target_sync = target_sync
        self.updates = 0
        self.eps = 1.0
        self.eps_min, self.eps_decay = 0.01, 0.995
        self.n_act = n_act
        # ring replay buffer, device-resident
        self.cap = buffer_size
        self.size = 0
        self.ptr = 0
        self.b_obs = torch.zeros(buffer_size, n_obs, device=device)
        self.b_act = torch.zeros(buffer_size, dtype=torch.long,
                                 device=device)
        self.b_rew = torch.zeros(buffer_size, device=device)
        self.b_next = torch.zeros(buffer_size, n_obs, device=device)
        self.b_done = torch.zeros(buffer_size, device=device)

    def act(self, obs: torch.Tensor) -> torch.Tensor:
        E = obs.shape[0]
        with torch.no_grad():
            greedy = self.q(obs).argmax(dim=1)
        explore = torch.rand(E, device=obs.device) < self.eps
        rand_a = torch.randint(0, self.n_act, (E,), device=obs.device)
        return torch.where(explore, rand_a, greedy)

    def remember(self, obs, act, rew, next_obs, done):
        n = obs.shape[0]
        idx = (self.ptr + torch.arange(n, device=obs.device)) % self.cap
        self.b_obs[idx] = obs
        self.b_act[idx] = act.long()
        self.b_rew[idx] = rew
        self.b_next[idx] = next_obs
        self.b_done[idx] = done
        self.ptr = (self.ptr + n) % self.cap
        self.size = min(self.size + n, self.cap)

    def replay(self) -> float:
        if self.size < self.batch:
            return 0.0
        idx = torch.randint(0, self.size, (self.batch,), device=self.device)
        obs, act = self.b_obs[idx], self.b_act[idx]
        rew, nxt, done = self.b_rew[idx], self.b_next[idx], self.b_done[idx]
        with torch.no_grad():
            tgt = rew + self.gamma * (1 - done) * self.target(nxt).max(1).values
        q = self.q(obs).gather(1, act[:, None]).squeeze(1)
        loss = nn.functional.smooth_l1_loss(q, tgt)
        self.opt.zero_grad()
        loss.backward()
        self.opt.step()
        self.updates += 1
        if self.updates % self.target_sync == 0:
            self.target.load_state_dict(self.q.state_dict())
        self.eps = max(self.eps_min, self.eps * self.eps_decay)
        return float(loss.detach())


class ActorCritic(nn.Module):
    def __init__(self, n_obs=N_OBS, n_act=N_ACT, width=64):
        super().__init__()
        self.body = nn.Sequential(
            nn.Linear(n_obs, width), nn.Tanh(),
            nn.Linear(width, width), nn.Tanh(),
        )
        self.pi = nn.Linear(width, n_act)
        self.v = nn.Linear(width, 1)

    def forward(self, x):
        h = self.body(x)
        return self.pi(h), self.v(h).squeeze(-1)


def gae_gpu(rewards, values, dones, gamma=0.99, lam=0.95):
    """(T,E) rewards/dones, (T+1,E) values -> (adv, returns) via the HIP
    reverse-scan kernel."""
    ops = require_hip_ops()
    T, E = rewards.shape
    adv = torch.empty_like(rewards)
    ret = torch.empty_like(rewards)
    stream = torch.cuda.current_stream(rewards.device).cuda_stream
    ops.gae(rewards.contiguous().data_ptr(), values.contiguous().data_ptr(),
            dones.contiguous().data_ptr(), adv.data_ptr(), ret.data_ptr(),
            T, E, gamma, lam, stream)
    return adv, ret


def gae_reference(rewards, values, dones, gamma=0.99, lam=0.95):
    """Plain torch fp32 reference for the GAE kernel."""
    T, E = rewards.shape
    adv = torch.zeros_like(rewards)
    a = torch.zeros(E, dtype=rewards.dtype, device=rewards.device)
    for t in reversed(range(T)):
        nonterm = 1.0 - dones[t]
        delta = rewards[t] + gamma * values[t + 1] * nonterm - values[t]
        a = delta + gamma * lam * nonterm * a
        adv[t] = a
    return adv, adv + values[:-1]


class PPOAgent:
    """Clipped-objective PPO over the vectorized HIP envs; gradients
    all-reduced across ranks (DP over RCCL) after each minibatch."""

    def __init__(self, device, n_obs=N_OBS, n_act=N_ACT, lr=3e-4,
                 gamma=0.99, lam=0.95, clip=0.2, epochs=4,
                 minibatches=4, ent_coef=0.01, vf_coef=0.5, seed=0,
                 use_graph: bool = False):
        torch.manual_seed(seed)
        self.device = torch.device(device)
        self.net = ActorCritic(n_obs, n_act).to(device)
        # capturable Adam: the optimizer step must be hip-graph-replayable
        # when use_graph captures the whole update phase
        self.opt = torch.optim.Adam(self.net.parameters(), lr=lr,
                                    capturable=bool(use_graph) and
                                    self.device.type == "cuda")
        self.gamma, self.lam, self.clip = gamma, lam, clip
        self.epochs, self.minibatches = epochs, minibatches
        self.ent_coef, self.vf_coef = ent_coef, vf_coef
        # hipGraph-captured rollout: the rollout is launch-bound (T
        # alternating tiny policy/env kernels), so one graph replay per
        # rollout removes the per-launch overhead. Capture is lazy and
        # per (env, T); falls back to eager on capture failure.
        self.use_graph = use_graph
        self._graph = None
        self._graph_key = None
        self._gbuf = None
        self._ugraph = None
        self._ugraph_key = None
        self._uloss = None
        # DDP gradient path: the eager update uses bucketed async
        # all-reduce overlapped with backward (parallel/dist.py
        # GradBucketer); the hip-graphed update keeps the capturable
        # synchronous flat collective (_allreduce_grads) because hook-
        # launched async ops cannot be graph-captured. Hooks register
        # permanently, so the bucketer exists only in eager+dist mode.
        self._bucketer = None
        if pdist.is_dist() and not (use_graph and
                                    self.device.type == "cuda"):
            self._bucketer = pdist.GradBucketer(self.net.parameters())

    @torch.no_grad()
    def policy(self, obs):
        logits, v = self.net(obs)
        dist_ = torch.distributions.Categorical(logits=logits)
        a = dist_.sample()
        return a, dist_.log_prob(a), v

    def _rollout_body(self, env, T, bufs):
        """The rollout loop against preallocated buffers — runs eagerly
        AND under hip-graph stream capture (no host reads, no allocs that
        escape; sampling via gumbel-argmax on capture-safe torch.rand)."""
        obs_buf, act_buf, logp_buf, rew_buf, done_buf, val_buf = bufs
        obs = env.obs
        for t in range(T):
            logits, v = self.net(obs)
            logp_all = torch.log_softmax(logits, dim=-1)
            u = torch.rand_like(logits)
            g = -torch.log(-torch.log(u + 1e-20) + 1e-20)
            a = (logp_all + g).argmax(dim=-1)
            obs_buf[t].copy_(obs)
            act_buf[t].copy_(a)
            logp_buf[t].copy_(logp_all.gather(1, a[:, None]).squeeze(1))
            val_buf[t].copy_(v)
            o, r, d = env.step(a)
            rew_buf[t].copy_(r)
            done_buf[t].copy_(d)
            obs = env.obs
        _, v_last = self.net(obs)
        val_buf[T].copy_(v_last)

    def _make_bufs(self, env, T):
        E, dev = env.n_envs, self.device
        return (torch.zeros(T, E, N_OBS, device=dev),
                torch.zeros(T, E, dtype=torch.long, device=dev),
                torch.zeros(T, E, device=dev),
                torch.zeros(T, E, device=dev),
                torch.zeros(T, E, device=dev),
                torch.zeros(T + 1, E, device=dev))

    @torch.no_grad()
    def rollout_graphed(self, env: TradingVecEnv, T: int):
        key = (id(env), T)
        if self._graph_key != key:
            self._gbuf = self._make_bufs(env, T)
            torch.cuda.synchronize()
            # warmup on a side stream (required before capture)
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                self._rollout_body(env, T, self._gbuf)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._rollout_body(env, T, self._gbuf)
            self._graph = g
            self._graph_key = key
        else:
            self._graph.replay()
        return self._gbuf

    def rollout(self, env: TradingVecEnv, T: int):
        E = env.n_envs
        dev = self.device
        obs_buf = torch.zeros(T, E, N_OBS, device=dev)
        act_buf = torch.zeros(T, E, dtype=torch.long, device=dev)
        logp_buf = torch.zeros(T, E, device=dev)
        rew_buf = torch.zeros(T, E, device=dev)
        done_buf = torch.zeros(T, E, device=dev)
        val_buf = torch.zeros(T + 1, E, device=dev)
        obs = env.obs
        for t in range(T):
            a, logp, v = self.policy(obs)
            obs_buf[t] = obs
            act_buf[t], logp_buf[t], val_buf[t] = a, logp, v
            obs, rew, done = env.step(a)
            rew_buf[t] = rew
            done_buf[t] = done
        with torch.no_grad():
            _, v_last = self.net(obs)
        val_buf[T] = v_last
        return obs_buf, act_buf, logp_buf, rew_buf, done_buf, val_buf

    def update(self, obs, act, logp_old, rew, done, val) -> dict:
        use_hip = obs.is_cuda
        if use_hip:
            adv, ret = gae_gpu(rew, val, done, self.gamma, self.lam)
        else:
            adv, ret = gae_reference(rew, val, done, self.gamma, self.lam)
        T, E = rew.shape
        n = T * E
        obs_f = obs.reshape(n, -1)
        act_f = act.reshape(n)
        logp_f = logp_old.reshape(n)
        adv_f = adv.reshape(n)
        ret_f = ret.reshape(n)
        adv_f = (adv_f - adv_f.mean()) / (adv_f.std() + 1e-8)
        stats = {"pi_loss": 0.0, "v_loss": 0.0, "entropy": 0.0}
        mb = n // self.minibatches
        for _ in range(self.epochs):
            perm = torch.randperm(n, device=obs.device)
            for i in range(self.minibatches):
                j = perm[i * mb:(i + 1) * mb]
                logits, v = self.net(obs_f[j])
                dist_ = torch.distributions.Categorical(logits=logits)
                logp = dist_.log_prob(act_f[j])
                ratio = torch.exp(logp - logp_f[j])
                s1 = ratio * adv_f[j]
                s2 = torch.clamp(ratio, 1 - self.clip, 1 + self.clip) * adv_f[j]
                pi_loss = -torch.min(s1, s2).mean()
                v_loss = ((v - ret_f[j]) ** 2).mean()
                ent = dist_.entropy().mean()
                loss = pi_loss + self.vf_coef * v_loss - self.ent_coef * ent
                self.opt.zero_grad()
                loss.backward()          # bucketer hooks overlap comm
                if self._bucketer is not None:
                    self._bucketer.finalize()
                else:
                    self._allreduce_grads()
                self.opt.step()
                stats["pi_loss"] += float(pi_loss.detach())
                stats["v_loss"] += float(v_loss.detach())
                stats["entropy"] += float(ent.detach())
        k = self.epochs * self.minibatches
        return {k2: v2 / k for k2, v2 in stats.items()}

    def _allreduce_grads(self):
        """DP gradient all-reduce (flat single bucket — the net is tiny, so
        one fused collective beats per-tensor latency on xGMI)."""
        if not pdist.is_dist():
            return
        import torch.distributed as dist
        grads = [p.grad for p in self.net.parameters() if p.grad is not None]
        flat = torch.cat([g.reshape(-1) for g in grads])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
        flat /= dist.get_world_size()
        off = 0
        for g in grads:
            g.copy_(flat[off:off + g.numel()].view_as(g))
            off += g.numel()

    def _update_body(self, bufs, loss_out):
        """The PPO update against the static rollout buffers — runs
        eagerly AND under hip-graph capture (no host reads; losses
        accumulate into a device tensor)."""
        obs, act, logp_old, rew, done, val = bufs
        adv, ret = gae_gpu(rew, val, done, self.gamma, self.lam)
        T, E = rew.shape
        n = T * E
        obs_f = obs.reshape(n, -1)
        act_f = act.reshape(n)
        logp_f = logp_old.reshape(n)
        adv_f = adv.reshape(n)
        ret_f = ret.reshape(n)
        adv_f = (adv_f - adv_f.mean()) / (adv_f.std() + 1e-8)
        loss_out.zero_()
        mb = n // self.minibatches
        for _ in range(self.epochs):
            perm = torch.randperm(n, device=obs.device)
            for i in range(self.minibatches):
                j = perm[i * mb:(i + 1) * mb]
                logits, v = self.net(obs_f[j])
                # manual categorical math: torch.distributions validates
                # with a host sync, which hip-graph capture forbids
                logp_all = torch.log_softmax(logits, dim=-1)
                logp = logp_all.gather(1, act_f[j][:, None]).squeeze(1)
                ratio = torch.exp(logp - logp_f[j])
                s1 = ratio * adv_f[j]
                s2 = torch.clamp(ratio, 1 - self.clip,
                                 1 + self.clip) * adv_f[j]
                pi_loss = -torch.min(s1, s2).mean()
                v_loss = ((v - ret_f[j]) ** 2).mean()
                ent = -(logp_all.exp() * logp_all).sum(-1).mean()
                loss = pi_loss + self.vf_coef * v_loss \
                    - self.ent_coef * ent
                self.opt.zero_grad(set_to_none=False)
                loss.backward()
                self._allreduce_grads()
                self.opt.step()
                loss_out[0] += pi_loss.detach()
                loss_out[1] += v_loss.detach()
                loss_out[2] += ent.detach()

    def update_graphed(self, bufs) -> dict:
        """Whole update phase as one hipGraph (forward+backward+capturable
        Adam, the torch full-iteration-capture pattern)."""
        key = tuple(id(b) for b in bufs)
        if self._ugraph_key != key:
            self._uloss = torch.zeros(3, device=self.device)
            # eager warmup initializes Adam state + grad buffers
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                self._update_body(bufs, self._uloss)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._update_body(bufs, self._uloss)
            self._ugraph = g
            self._ugraph_key = key
        else:
            self._ugraph.replay()
        k = self.epochs * self.minibatches
        vals = (self._uloss / k).cpu()
        return {"pi_loss": float(vals[0]), "v_loss": float(vals[1]),
                "entropy": float(vals[2])}

    def train_step(self, env: TradingVecEnv, horizon: int = 128) -> dict:
        if self.use_graph and self.device.type == "cuda":
            try:
                out = self.rollout_graphed(env, horizon)
            except Exception:
                self.use_graph = False        # fall back to eager forever
                out = self.rollout(env, horizon)
            if self.use_graph:
                try:
                    return self.update_graphed(out)
                except Exception:
                    self.use_graph = False
        else:
            out = self.rollout(env, horizon)
        return self.update(*out)
