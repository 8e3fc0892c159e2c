"""A3C actor-critic network for the rl workload
(reference workloads/pytorch/rl: Pong conv actor-critic)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class ActorCritic(nn.Module):
    def __init__(self, num_inputs=1, num_actions=6):
        super().__init__()
        self.conv1 = nn.Conv2d(num_inputs, 32, 5, 1, 2)
        self.conv2 = nn.Conv2d(32, 32, 5, 1, 1)
        self.conv3 = nn.Conv2d(32, 64, 4, 1, 1)
        self.conv4 = nn.Conv2d(64, 64, 3, 1, 1)
        self.lstm = nn.LSTMCell(1024, 512)
        self.critic = nn.Linear(512, 1)
        self.actor = nn.Linear(512, num_actions)

    def forward(self, x, hx, cx):
        x = F.relu(F.max_pool2d(self.conv1(x), 2, 2))
        x = F.relu(F.max_pool2d(self.conv2(x), 2, 2))
        x = F.relu(F.max_pool2d(self.conv3(x), 2, 2))
        x = F.relu(F.max_pool2d(self.conv4(x), 2, 2))
        x = x.view(x.size(0), -1)
        hx, cx = self.lstm(x, (hx, cx))
        return self.critic(hx), self.actor(hx), hx, cx
