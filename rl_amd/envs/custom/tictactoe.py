"""TicTacToeEnv — turn-based two-player board env.

Reference: pytorch/rl torchrl/envs/custom/ (TicTacToeEnv): 3×3 board,
players alternate under ``turn``; ``mask`` marks legal moves; win = +1
to the winner (observed on the winning player's transition), illegal
moves lose immediately when ``single_player=False``.
"""
from __future__ import annotations

from typing import Optional

import torch

from ...data.tensor_specs import Binary, Categorical, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["TicTacToeEnv"]

_WIN_LINES = torch.tensor(
    [
        [0, 1, 2], [3, 4, 5], [6, 7, 8],  # rows
        [0, 3, 6], [1, 4, 7], [2, 5, 8],  # cols
        [0, 4, 8], [2, 4, 6],             # diagonals
    ]
)


class TicTacToeEnv(EnvBase):
    def __init__(self, *, single_player: bool = False, device=None):
        super().__init__(device=device, batch_size=())
        self.single_player = single_player
        self.observation_spec = Composite(
            {
                "board": Unbounded(shape=(3, 3), dtype=torch.int8, device=self.device),
                "turn": Unbounded(shape=(1,), dtype=torch.int8, device=self.device),
                "mask": Binary(shape=(9,), device=self.device),
            },
            shape=(),
            device=self.device,
        )
        self.action_spec = Categorical(9, shape=(), device=self.device)
        self.reward_spec = Composite(
            {
                ("player0", "reward"): Unbounded(shape=(1,), device=self.device),
                ("player1", "reward"): Unbounded(shape=(1,), device=self.device),
            },
            shape=(),
            device=self.device,
        )
        self._board: Optional[torch.Tensor] = None
        self._turn = 0
        self._gen = torch.Generator(device="cpu")

    def _obs_td(self, r0=0.0, r1=0.0, done=False):
        td = TensorDict({}, batch_size=(), device=self.device)
        td.set("board", self._board.reshape(3, 3).clone())
        td.set("turn", torch.tensor([self._turn], dtype=torch.int8, device=self.device))
        td.set("mask", (self._board == 0).clone())
        td.set(("player0", "reward"), torch.tensor([r0], device=self.device))
        td.set(("player1", "reward"), torch.tensor([r1], device=self.device))
        td.set("reward", torch.tensor([r0 - r1], device=self.device))
        done_t = torch.tensor([done], device=self.device)
        td.set("done", done_t)
        td.set("terminated", done_t.clone())
        return td

    def _reset(self, tensordict=None, **kwargs) -> TensorDictBase:
        self._board = torch.zeros(9, dtype=torch.int8, device=self.device)
        self._turn = 0
        td = self._obs_td()
        td.pop(("player0", "reward"), None)
        td.pop(("player1", "reward"), None)
        td.pop("reward", None)
        return td

    def _winner(self) -> int:
        """0/1 = winning player, -1 = none."""
        for player, mark in ((0, 1), (1, -1)):
            lines = self._board[_WIN_LINES.to(self.device)]
            if bool(((lines == mark).all(-1)).any()):
                return player
        return -1

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        action = int(tensordict.get("action").item())
        player = self._turn
        mark = 1 if player == 0 else -1
        if self._board[action] != 0:
            # illegal: immediate loss for the mover
            r0, r1 = (-1.0, 0.0) if player == 0 else (0.0, -1.0)
            return self._obs_td(r0, r1, done=True)
        self._board[action] = mark
        winner = self._winner()
        full = bool((self._board != 0).all())
        done = winner >= 0 or full
        r0 = 1.0 if winner == 0 else (-1.0 if winner == 1 else 0.0)
        r1 = 1.0 if winner == 1 else (-1.0 if winner == 0 else 0.0)
        self._turn = 1 - self._turn
        td = self._obs_td(r0, r1, done)
        if self.single_player and not done and self._turn == 1:
            # scripted random opponent
            legal = (self._board == 0).nonzero().reshape(-1)
            opp = legal[torch.randint(0, legal.numel(), (1,), generator=self._gen)]
            self._board[opp] = -1
            winner = self._winner()
            full = bool((self._board != 0).all())
            done = winner >= 0 or full
            r0 = 1.0 if winner == 0 else (-1.0 if winner == 1 else 0.0)
            r1 = 1.0 if winner == 1 else (-1.0 if winner == 0 else 0.0)
            self._turn = 0
            td = self._obs_td(r0, r1, done)
        return td

    def _set_seed(self, seed):
        if seed is not None:
            self._gen.manual_seed(seed)
        return seed
