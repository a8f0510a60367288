from .comm import (
    GradAllReducer,
    all_reduce_grads,
    broadcast_tensordict,
    init_distributed,
    irecv_tensordict,
    isend_tensordict,
    recv_tensordict,
    rendezvous_store,
    send_tensordict,
)
from .mailbox import CommandChannel, Mailbox, RequestReply
from .replay_service import ReplayBufferClient, ReplayBufferService
