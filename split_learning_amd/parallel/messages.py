"""Wire-level message types.

Control-plane messages are plain dicts with an "action" key, matching the
reference protocol verbs (REGISTER/START/SYN/NOTIFY/PAUSE/UPDATE/STOP —
reference src/Server.py:103-212, src/RpcClient.py:43-135).

Data-plane messages carry GPU-resident tensors: the loopback plane passes them
by reference; the RCCL plane moves them as p2p sends over xGMI (replacing the
reference's pickle-numpy-over-AMQP payloads, src/train/VGG16.py:26-47).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

import torch


@dataclass
class ActivationMsg:
    data_id: int
    data: torch.Tensor
    labels: Optional[torch.Tensor]
    # sender-id stack for gradient routing (reference "trace",
    # src/train/VGG16.py:24-31: push on forward, pop on backward)
    trace: List[int] = field(default_factory=list)


@dataclass
class GradientMsg:
    data_id: int
    data: torch.Tensor
    trace: List[int] = field(default_factory=list)
