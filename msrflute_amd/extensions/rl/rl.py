"""RL-based aggregation-weight estimation for DGA (arXiv:2106.07578).

Reference: extensions/RL/RL.py:79-343.  The shipped reference RL path is
broken (calls ``.np()``, references attributes that don't exist —
SURVEY.md §7.5), so this is a reimplementation of the documented intent:
a DQN whose state is the concatenation of the round's client weights and
gradient statistics (mag/mean/var), whose action is a per-client log-weight
vector, trained from a replay memory against rewards derived from the
validation-accuracy delta of the RL-weighted vs. softmax-weighted model.

Determinism: the network init and replay sampling are seeded so the
symmetric rank replicas evolve identically (every rank runs the same RL
computation on the same gathered stats).
"""

from __future__ import annotations

import os
import random
from collections import deque

import numpy as np
import torch
import torch.nn as nn

from ...utils import print_rank, to_device


class NeuralNetwork(nn.Module):
    """MLP Q-network (reference: RL.py:79-116)."""

    def __init__(self, state_dim, hidden_dim, out_dim):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(state_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, out_dim),
        )

    def forward(self, x):
        return self.net(x)


class BatchRNN(nn.Module):
    """BiLSTM Q-network over the state sequence (reference: RL.py:118-145)."""

    def __init__(self, state_dim, hidden_dim, out_dim):
        super().__init__()
        self.rnn = nn.LSTM(input_size=1, hidden_size=hidden_dim,
                           bidirectional=True, batch_first=True)
        self.fc = nn.Linear(2 * hidden_dim, out_dim)
        self.state_dim = state_dim

    def forward(self, x):
        seq = x.view(-1, x.shape[-1], 1)
        out, _ = self.rnn(seq)
        return self.fc(out[:, -1, :])


class RL:
    """DQN weight estimator (reference: RL.py:149-343)."""

    def __init__(self, config=None, seed: int = 12345):
        rl_config = (config or {}).get("RL", {}) or {}
        self.config = rl_config
        self.gamma = rl_config.get("gamma", 0.99)
        self.epsilon = rl_config.get("epsilon", 0.5)
        self.epsilon_decay = rl_config.get("epsilon_decay", 0.99)
        self.final_epsilon = rl_config.get("final_epsilon", 0.01)
        self.hidden_dim = rl_config.get("hidden_dim", 512)
        self.lr = rl_config.get("lr", 0.001)
        self.batch_size = rl_config.get("batch_size", 16)
        self.memory_size = rl_config.get("memory_size", 1000)
        self.network_type = rl_config.get("network_type", "mlp")
        self.out_dim = rl_config.get("num_clients", None)
        self.model_path = rl_config.get("RL_path", "rl_model")
        self.running_loss = 0.0
        self.runningLoss = 0.0  # reference attribute name
        self.rl_weights = None
        self.rl_losses = (None, None)
        self.memory = deque(maxlen=self.memory_size)
        self.model = None
        self.optimizer = None
        self._rng = random.Random(seed)
        self._seed = seed

    def _ensure_model(self, state_dim, out_dim):
        if self.model is None:
            torch.manual_seed(self._seed)
            cls = BatchRNN if self.network_type.lower() in ("rnn", "lstm", "bilstm") \
                else NeuralNetwork
            self.model = to_device(cls(state_dim, self.hidden_dim, out_dim))
            self.optimizer = torch.optim.Adam(self.model.parameters(), lr=self.lr)
            self._state_dim = state_dim
            self._out_dim = out_dim

    def forward(self, state):
        """ε-greedy action: per-client log-weights (reference: RL.py:185-204)."""
        state = np.asarray(state, dtype=np.float32)
        out_dim = self.out_dim or (len(state) // 4)
        self._ensure_model(len(state), out_dim)
        if self._rng.random() < self.epsilon:
            self.epsilon = max(self.epsilon * self.epsilon_decay,
                               self.final_epsilon)
            action = np.array([self._rng.uniform(-1, 1) for _ in range(out_dim)],
                              dtype=np.float32)
            return action
        self.epsilon = max(self.epsilon * self.epsilon_decay,
                           self.final_epsilon)
        with torch.no_grad():
            t = to_device(torch.from_numpy(state)).unsqueeze(0)
            q = self.model(t).squeeze(0).detach().cpu().numpy()
        return q

    def set_weights(self, weights):
        self.rl_weights = np.asarray(weights, dtype=np.float64)

    def set_losses(self, losses):
        self.rl_losses = losses

    def train(self, batch):
        """One replay step (reference: RL.py:206-262): push the round's
        (state, action, reward) transition, then fit the DQN on a sampled
        minibatch with ``q = Q(s)·a`` toward the γ-discounted target
        ``y = r + γ·max Q(s')``.

        The reference computes exactly ``q = Σ Q(s)·a`` and ``y = r`` —
        its own comment ("set y_j to r_j for terminal state, otherwise to
        r_j + gamma*max(Q)") documents the discounted form but every FL
        round was treated as terminal.  Rounds are NOT terminal (the next
        round's state is the successor), so this implements the
        documented intent: each transition's next state is backfilled
        when the following round arrives, and transitions still awaiting
        a successor use y = r.  γ=0 reproduces the reference's update
        exactly."""
        state, action, reward = batch
        state = np.asarray(state, dtype=np.float32)
        action = np.asarray(action, dtype=np.float32)
        r = float(np.asarray(reward).reshape(-1)[0])
        # backfill the previous transition's successor state
        if self.memory:
            s_prev, a_prev, r_prev, _ = self.memory[-1]
            self.memory[-1] = (s_prev, a_prev, r_prev, state)
        self.memory.append((state, action, r, None))
        self._ensure_model(len(state), len(action))

        n = min(self.batch_size, len(self.memory))
        sample = self._rng.sample(list(self.memory), n)
        # pad/truncate stored episodes to the current dims (cohort size
        # varies round to round)
        S = np.zeros((n, self._state_dim), dtype=np.float32)
        A = np.zeros((n, self._out_dim), dtype=np.float32)
        R = np.zeros(n, dtype=np.float32)
        SN = np.zeros((n, self._state_dim), dtype=np.float32)
        has_next = np.zeros(n, dtype=np.float32)
        for i, (s, a, rr, sn) in enumerate(sample):
            S[i, :min(len(s), self._state_dim)] = s[:self._state_dim]
            A[i, :min(len(a), self._out_dim)] = a[:self._out_dim]
            R[i] = rr
            if sn is not None:
                SN[i, :min(len(sn), self._state_dim)] = sn[:self._state_dim]
                has_next[i] = 1.0
        S_t = to_device(torch.from_numpy(S))
        A_t = to_device(torch.from_numpy(A))
        R_t = to_device(torch.from_numpy(R))
        self.model.train()
        with torch.no_grad():
            next_max = self.model(to_device(torch.from_numpy(SN))).max(dim=1).values
        y = (R_t + self.gamma * to_device(torch.from_numpy(has_next))
             * next_max).detach()
        q = (self.model(S_t) * A_t).sum(dim=1)
        self.optimizer.zero_grad()
        loss = torch.nn.functional.mse_loss(q, y)
        loss.backward()
        self.optimizer.step()
        # running EMA like the reference (RL.py:258-262)
        cur = float(loss.item())
        self.running_loss = cur if self.running_loss == 0.0 \
            else 0.95 * self.running_loss + 0.05 * cur
        self.runningLoss = self.running_loss
        return self.running_loss

    def save(self, curr_iter=None):
        """Independent RL checkpoint (reference: RL.py:296-343)."""
        if self.model is None:
            return
        os.makedirs(self.model_path, exist_ok=True)
        path = os.path.join(self.model_path, "rl_model.tar")
        torch.save({
            "model_state_dict": self.model.state_dict(),
            "optimizer_state_dict": self.optimizer.state_dict(),
            "epsilon": self.epsilon,
            "iter": curr_iter,
            "state_dim": self._state_dim,
            "out_dim": self._out_dim,
        }, path)
        print_rank(f"saved RL model to {path}")

    def load(self):
        path = os.path.join(self.model_path, "rl_model.tar")
        if not os.path.isfile(path):
            return False
        ckpt = torch.load(path, map_location="cpu", weights_only=False)
        self._ensure_model(ckpt["state_dim"], ckpt["out_dim"])
        self.model.load_state_dict(ckpt["model_state_dict"])
        self.optimizer.load_state_dict(ckpt["optimizer_state_dict"])
        self.epsilon = ckpt.get("epsilon", self.epsilon)
        return True
