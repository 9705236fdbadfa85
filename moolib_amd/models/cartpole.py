"""Small MLP (+optional LSTMCell) policy for CartPole-class env vectors.

Capability parity with the reference's examples/a2c.py Model; used by the
A2C example and the end-to-end learning test.
"""
import torch
import torch.nn.functional as F
from torch import nn


class CartPoleNet(nn.Module):
    def __init__(self, obs_dim=4, num_actions=2, use_lstm=True, hidden=64):
        super().__init__()
        self.use_lstm = use_lstm
        self.fc0 = nn.Linear(obs_dim, 128)
        self.fc1 = nn.Linear(128, hidden)
        if use_lstm:
            self.core = nn.LSTMCell(hidden, hidden)
        self.policy = nn.Linear(hidden, num_actions)
        self.baseline = nn.Linear(hidden, 1)
        self.hidden = hidden

    def initial_state(self, batch_size=1):
        if not self.use_lstm:
            return tuple()
        return tuple(torch.zeros(batch_size, self.hidden) for _ in range(2))

    def forward(self, observation, done, core_state, unroll=False):
        if not unroll:
            observation = observation.unsqueeze(0)
            done = done.unsqueeze(0)
        T, B = observation.shape[:2]
        x = observation.reshape(T * B, -1)
        x = torch.tanh(self.fc0(x))
        x = torch.tanh(self.fc1(x))

        if self.use_lstm:
            x = x.view(T, B, -1)
            notdone = (~done).float().unsqueeze(-1)
            outs = []
            for t in range(T):
                core_state = tuple(notdone[t] * s for s in core_state)
                h, c = self.core(x[t], core_state)
                core_state = (h, c)
                outs.append(h)
            core_output = torch.cat(outs)
        else:
            core_output = x.view(T * B, -1) if x.dim() > 2 else x

        logits = self.policy(core_output)
        baseline = self.baseline(core_output)
        action = torch.multinomial(F.softmax(logits, dim=-1), num_samples=1)

        action = action.view(T, B)
        logits = logits.view(T, B, -1)
        baseline = baseline.view(T, B)
        if not unroll:
            action, logits, baseline = action[0], logits[0], baseline[0]
        return (action, logits, baseline), core_state
