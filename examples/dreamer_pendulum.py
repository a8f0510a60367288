"""Dreamer-style model-based RL on the torch-native Pendulum
(reference sota-implementations/dreamer/dreamer.py shape, compact).

Three phases per iteration: (1) collect real transitions, (2) train the
RSSM world model (reconstruction + reward + KL), (3) train actor/value
in imagination through the world model.

Run: python examples/dreamer_pendulum.py [--iters 5]
"""
from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rl_amd.collectors import Collector
from rl_amd.data import LazyTensorStorage, SliceSampler, TensorDictReplayBuffer
from rl_amd.envs.custom import PendulumEnv
from rl_amd.modules import MLP, NormalParamExtractor, ProbabilisticActor, TanhNormal
from rl_amd.modules.models.model_based import ObsDecoder, ObsEncoder, RSSMPosterior, RSSMPrior
from rl_amd.tensordict import TensorDict, TensorDictModule


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=5)
    p.add_argument("--horizon", type=int, default=5, help="imagination horizon")
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    env = PendulumEnv(batch_size=[8], device=device)
    obs_dim = env.observation_spec["observation"].shape[-1]
    act_dim = env.action_spec.shape[-1]
    state_dim, rnn_dim = 16, 32

    encoder = MLP(in_features=obs_dim, out_features=32, num_cells=[64], device=device)
    prior = RSSMPrior(
        action_dim=act_dim, stoch_dim=state_dim, deter_dim=rnn_dim,
        hidden_dim=64, device=device,
    )
    posterior = RSSMPosterior(
        deter_dim=rnn_dim, embed_dim=32, stoch_dim=state_dim,
        hidden_dim=64, device=device,
    )
    decoder = MLP(
        in_features=state_dim + rnn_dim, out_features=obs_dim, num_cells=[64], device=device
    )
    reward_model = MLP(
        in_features=state_dim + rnn_dim, out_features=1, num_cells=[64], device=device
    )
    wm_params = (
        list(encoder.parameters()) + list(prior.parameters())
        + list(posterior.parameters()) + list(decoder.parameters())
        + list(reward_model.parameters())
    )
    wm_optim = torch.optim.Adam(wm_params, lr=3e-4)

    actor = ProbabilisticActor(
        TensorDictModule(
            torch.nn.Sequential(
                MLP(in_features=state_dim + rnn_dim, out_features=2 * act_dim,
                    num_cells=[64], device=device),
                NormalParamExtractor(),
            ),
            in_keys=["latent"],
            out_keys=["loc", "scale"],
        ),
        in_keys=["loc", "scale"],
        distribution_class=TanhNormal,
        return_log_prob=False,
    )
    value = MLP(in_features=state_dim + rnn_dim, out_features=1, num_cells=[64], device=device)
    ac_optim = torch.optim.Adam(
        list(actor.parameters()) + list(value.parameters()), lr=3e-4
    )

    T = 12
    rb = TensorDictReplayBuffer(
        storage=LazyTensorStorage(20_000, device=device),
        sampler=SliceSampler(slice_len=T),
        batch_size=4 * T,
    )
    col = Collector(env, None, frames_per_batch=8 * 25, total_frames=-1, use_graph=False)
    col_iter = iter(col)

    def world_model_loss(seq):
        """seq: [B, T] real transitions → ELBO-ish losses."""
        B = seq.batch_size[0]
        h = torch.zeros(B, rnn_dim, device=device)
        s = torch.zeros(B, state_dim, device=device)
        rec_loss = rew_loss = kl_loss = 0.0
        for t in range(T):
            obs_emb = encoder(seq.get("observation")[:, t])
            act = seq.get("action")[:, t]
            _sp, prior_mean, prior_std, h = prior(s, h, act)
            s, post_mean, post_std = posterior(h, obs_emb)
            latent = torch.cat([s, h], -1)
            rec = decoder(latent)
            rec_loss = rec_loss + (rec - seq.get(("next", "observation"))[:, t]).pow(2).mean()
            rew_hat = reward_model(latent)
            rew_loss = rew_loss + (rew_hat - seq.get(("next", "reward"))[:, t]).pow(2).mean()
            kl = (
                (prior_std.log() - post_std.log())
                + (post_std.pow(2) + (post_mean - prior_mean).pow(2))
                / (2 * prior_std.pow(2))
                - 0.5
            ).mean()
            kl_loss = kl_loss + kl
        return rec_loss / T, rew_loss / T, kl_loss / T, (s.detach(), h.detach())

    for it in range(args.iters):
        batch = next(col_iter)
        rb.extend(batch.reshape(-1))
        if len(rb) < 4 * T:
            continue
        # --- world model phase ---
        seq = rb.sample().reshape(4, T)
        rec, rew, kl, (s0, h0) = world_model_loss(seq)
        wm_total = rec + rew + 0.1 * kl
        wm_optim.zero_grad()
        wm_total.backward()
        wm_optim.step()
        # --- imagination phase: dream from the final posterior state ---
        s, h = s0, h0
        returns = 0.0
        for _ in range(args.horizon):
            latent = torch.cat([s, h], -1)
            td = TensorDict({"latent": latent}, batch_size=[latent.shape[0]], device=device)
            act = actor(td).get("action")
            s, _m, _st, h = prior(s, h, act)
            returns = returns + reward_model(torch.cat([s, h], -1)).squeeze(-1)
        v_final = value(torch.cat([s, h], -1)).squeeze(-1)
        actor_loss = -(returns + v_final).mean()
        value_loss = (value(torch.cat([s0, h0], -1)).squeeze(-1) - (returns + v_final).detach()).pow(2).mean()
        ac_optim.zero_grad()
        (actor_loss + value_loss).backward()
        ac_optim.step()
        print(
            f"iter {it}: rec {float(rec):.4f} rew {float(rew):.4f} "
            f"kl {float(kl):.4f} dream_return {float(returns.mean()):.3f}"
        )
    col.shutdown()
    print("done")


if __name__ == "__main__":
    main()
