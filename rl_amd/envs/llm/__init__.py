from .chat import ChatEnv
