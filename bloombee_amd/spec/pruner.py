"""Mid-network speculative-tree pruning (parity: reference
server/speculative_pruner/ — SimpleProbabilityPruner over a MidLMHead
auxiliary vocab head at the LAST server block, plus the adaptive neural
pruner and the factory/enum config; ref pruner_manager.py:14-60,
adaptive_neural_pruner.py:15-519, mid_layer_LM_head.py:10-22)."""
from __future__ import annotations

import enum
from typing import List, Optional

import torch

from bloombee_amd import ops
from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)


class PruningMethod(enum.Enum):
    NONE = "none"
    PROBABILITY = "probability"
    ADAPTIVE = "adaptive"


class MidLMHead(torch.nn.Module):
    """Auxiliary vocab projection at a mid/last server block (ref
    mid_layer_LM_head.py:10-22). Random-init here (no pretrained head in this
    offline environment); optionally trained online from observed logits."""

    def __init__(self, hidden_size: int, vocab_size: int, seed: int = 0,
                 dtype=torch.bfloat16, device="cpu"):
        super().__init__()
        gen = torch.Generator().manual_seed(seed + 99)
        self.norm_w = torch.nn.Parameter(
            torch.ones(hidden_size, dtype=dtype, device=device),
            requires_grad=False)
        self.weight = torch.nn.Parameter(
            torch.randn(vocab_size, hidden_size, generator=gen).mul_(0.02)
            .to(dtype).to(device), requires_grad=False)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        return ops.linear(ops.rms_norm(hidden, self.norm_w), self.weight)


class SimpleProbabilityPruner:
    """Keep tree nodes whose token probability under the MidLMHead at their
    parent exceeds a threshold; ancestors of kept nodes are always kept
    (ref simple pruner, 241 LoC)."""

    def __init__(self, head: MidLMHead, threshold: float = 1e-3,
                 max_keep: Optional[int] = None):
        self.head = head
        self.threshold = threshold
        self.max_keep = max_keep

    @torch.no_grad()
    def keep_indices(self, hidden: torch.Tensor, tokens: List[int],
                     parents: List[int]) -> List[int]:
        """hidden: (T, H) tree-node hidden states (node i's output).
        Returns sorted linear indices to keep."""
        T = len(tokens)
        logits = self.head(hidden).float()
        probs = torch.softmax(logits, dim=-1)
        score = []
        for i in range(T):
            p = parents[i]
            if p == -1:
                score.append(1.0)  # first level always kept
            else:
                score.append(float(probs[p, tokens[i]]))
        keep = {i for i in range(T)
                if score[i] >= self.threshold or parents[i] == -1}
        # ancestors of kept nodes stay
        for i in list(keep):
            j = parents[i]
            while j != -1:
                keep.add(j)
                j = parents[j]
        kept = sorted(keep)
        if self.max_keep is not None and len(kept) > self.max_keep:
            ranked = sorted(kept, key=lambda i: -score[i])[: self.max_keep]
            keep = set()
            for i in ranked:
                keep.add(i)
                j = parents[i]
                while j != -1:
                    keep.add(j)
                    j = parents[j]
            kept = sorted(keep)
        return kept


class AdaptiveNeuralPruner(torch.nn.Module):
    """Small MLP over probability features deciding keep/drop per node, with
    online training against realized acceptance (ref adaptive_neural_pruner
    NodePruner, :15-519 — same feature set: parent prob, depth, sibling
    rank)."""

    def __init__(self, head: MidLMHead, hidden: int = 16, lr: float = 1e-2):
        super().__init__()
        self.head = head
        self.net = torch.nn.Sequential(
            torch.nn.Linear(3, hidden), torch.nn.ReLU(),
            torch.nn.Linear(hidden, 1))
        self.opt = torch.optim.SGD(self.net.parameters(), lr=lr)
        self._last_features: Optional[torch.Tensor] = None
        self._last_indices: Optional[List[int]] = None

    def _features(self, hidden, tokens, parents) -> torch.Tensor:
        with torch.no_grad():
            probs = torch.softmax(self.head(hidden).float(), -1)
        T = len(tokens)
        depth = [0] * T
        for i in range(T):
            depth[i] = 0 if parents[i] == -1 else depth[parents[i]] + 1
        feats = []
        for i in range(T):
            p = parents[i]
            pp = 1.0 if p == -1 else float(probs[p, tokens[i]])
            sib = [j for j in range(T) if parents[j] == parents[i]]
            rank = sorted(sib, key=lambda j: -(1.0 if parents[j] == -1 else
                                               float(probs[parents[j], tokens[j]]))
                          ).index(i)
            feats.append([pp, float(depth[i]), float(rank)])
        return torch.tensor(feats)

    @torch.no_grad()
    def keep_indices(self, hidden, tokens, parents) -> List[int]:
        feats = self._features(hidden, tokens, parents)
        scores = torch.sigmoid(self.net(feats).squeeze(-1))
        keep = {i for i in range(len(tokens)) if scores[i] > 0.5 or parents[i] == -1}
        for i in list(keep):
            j = parents[i]
            while j != -1:
                keep.add(j)
                j = parents[j]
        self._last_features = feats
        self._last_indices = sorted(keep)
        return self._last_indices

    def train_step(self, accepted: List[int]) -> float:
        """Online update: accepted node indices should score 1, rest 0."""
        if self._last_features is None:
            return 0.0
        target = torch.zeros(self._last_features.shape[0])
        for i in accepted:
            target[i] = 1.0
        self.opt.zero_grad()
        out = self.net(self._last_features).squeeze(-1)
        loss = torch.nn.functional.binary_cross_entropy_with_logits(out, target)
        loss.backward()
        self.opt.step()
        return float(loss.detach())


def create_pruner(method: PruningMethod, head: MidLMHead, **kw):
    """Factory (ref pruner_factory.py:14)."""
    if method == PruningMethod.NONE:
        return None
    if method == PruningMethod.PROBABILITY:
        return SimpleProbabilityPruner(head, **kw)
    if method == PruningMethod.ADAPTIVE:
        return AdaptiveNeuralPruner(head, **kw)
    raise ValueError(method)


def save_pruner(pruner, path: str) -> None:
    """Checkpoint a pruner's trained state (parity: reference
    speculative_pruner/lm_head_trainer.py:87 LM-head trainer checkpoints).
    Saves the mid-network head and, for the adaptive pruner, the scorer MLP
    + optimizer so online training resumes where it left off."""
    state = {"head": pruner.head.state_dict()}
    if isinstance(pruner, AdaptiveNeuralPruner):
        state["scorer"] = pruner.net.state_dict()
        state["opt"] = pruner.opt.state_dict()
    torch.save(state, path)


def load_pruner(pruner, path: str):
    """Restore state saved by save_pruner into a same-shape pruner."""
    state = torch.load(path, map_location="cpu", weights_only=False)
    pruner.head.load_state_dict(state["head"])
    if isinstance(pruner, AdaptiveNeuralPruner) and "scorer" in state:
        pruner.net.load_state_dict(state["scorer"])
        pruner.opt.load_state_dict(state["opt"])
    return pruner
