from .pendulum import PendulumEnv
from .chess import ChessEnv
from .synthetic import AntVec, HalfCheetahVec, HumanoidVec, SyntheticMuJoCoEnv
from .tictactoe import TicTacToeEnv
from .trading import TradingEnv
from .vla import ToyVLAEnv
