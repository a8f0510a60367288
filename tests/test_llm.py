"""LLM vertical tests: History, transformers wrapper, ChatEnv, collector,
GRPO/SFT — all on a tiny random-init model (no downloads)."""
import pytest
import torch

transformers = pytest.importorskip("transformers")

from rl_amd.collectors import LLMCollector
from rl_amd.data import History
from rl_amd.envs import ChatEnv
from rl_amd.modules import TransformersWrapper
from rl_amd.objectives import CISPOLoss, DAPO, GRPOLoss, MCAdvantage, SFTLoss
from rl_amd.tensordict import TensorDict
from rl_amd.testing.llm_mocks import ByteTokenizer, make_tiny_lm


@pytest.fixture(scope="module")
def lm():
    torch.manual_seed(0)
    return make_tiny_lm()


@pytest.fixture(scope="module")
def tok():
    return ByteTokenizer()


class TestHistory:
    def test_build_and_render(self):
        h = History.from_chats(
            [[{"role": "user", "content": "hi"}, {"role": "assistant", "content": "yo"}]]
        )
        text = h.apply_chat_template()
        assert "user: hi" in text[0]
        assert "assistant: yo" in text[0]

    def test_append_and_index(self):
        h = History(role="user", content="q1")
        h.append("assistant", "a1")
        assert h.last_role == "assistant"
        assert len(h) == 2

    def test_batched(self):
        h = History.from_text(["a", "b", "c"])
        assert h.batch_size == (3,)
        assert h[1].last_content == "b"

    def test_tensordict_roundtrip(self):
        h = History.from_text(["x", "y"])
        td = h.to_tensordict()
        h2 = History.from_tensordict(td)
        assert h2.contents == h.contents


class TestWrapper:
    def test_generate(self, lm, tok):
        wrap = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=5)
        h = History.from_text(["hello world"])
        td = TensorDict({}, batch_size=[1])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td = wrap(td)
        assert td.get("tokens_response").shape[1] == 5
        assert td.get("log_probs").shape == td.get("tokens_response").shape
        hist = td.get_non_tensor("history")
        assert hist["roles"][0][-1] == "assistant"

    def test_log_probs_mode_consistency(self, lm, tok):
        gen = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=4, do_sample=False)
        h = History.from_text(["abc"])
        td = TensorDict({}, batch_size=[1])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td = gen(td)
        # recompute log-probs of the greedy tokens with the SAME prompt:
        # drop the assistant turn the wrapper appended
        hist = td.get_non_tensor("history")
        td.set_non_tensor(
            "history",
            {"roles": [hist["roles"][0][:-1]], "contents": [hist["contents"][0][:-1]]},
        )
        lp_mod = TransformersWrapper(lm, tokenizer=tok, generate=False)
        td2 = lp_mod(td.clone(False))
        lp_gen = td.get("log_probs")
        lp_re = td2.get("log_probs")
        assert torch.allclose(lp_gen, lp_re, atol=1e-4), (lp_gen - lp_re).abs().max()


class TestChatEnv:
    def test_reset_step(self):
        env = ChatEnv(iter(["what is 2+2?"]), reward_fn=lambda h: float(len(h.last_content)))
        td = env.reset()
        hist = td.get_non_tensor("history")
        assert hist["roles"][0][-1] == "user"
        td.set_non_tensor("text_response", "4")
        td = env.step(td)
        assert td.get(("next", "reward")).item() == 1.0
        assert td.get(("next", "done")).item() is True

    def test_collector(self, lm, tok):
        env = ChatEnv(iter(["hi", "yo", "hey"]), reward_fn=lambda h: 1.0)
        policy = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=3)
        col = LLMCollector(env, policy, dialog_turns_per_batch=1, total_dialog_turns=2)
        batches = list(col)
        assert len(batches) == 2
        assert ("next", "reward") in batches[0].keys(True, True)


class TestGRPO:
    def _fake_rollout(self, lm, tok, n=4):
        gen = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=4)
        h = History.from_text(["q"] * n)
        td = TensorDict({}, batch_size=[n])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td = gen(td)
        # strip assistant turn so log-prob mode sees the same prompt
        hist = td.get_non_tensor("history")
        td.set_non_tensor(
            "history",
            {
                "roles": [r[:-1] for r in hist["roles"]],
                "contents": [c[:-1] for c in hist["contents"]],
            },
        )
        td.set("next", TensorDict({"reward": torch.randn(n, 1)}, batch_size=[n]))
        return td

    def test_mc_advantage(self, lm, tok):
        td = self._fake_rollout(lm, tok, n=4)
        MCAdvantage(grpo_size=2)(td)
        adv = td.get("advantage").reshape(-1, 2)
        assert torch.allclose(adv.mean(-1), torch.zeros(2), atol=1e-5)

    @pytest.mark.parametrize("loss_cls", [GRPOLoss, DAPO, CISPOLoss])
    def test_grpo_losses(self, lm, tok, loss_cls):
        td = self._fake_rollout(lm, tok)
        MCAdvantage(grpo_size=2)(td)
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        loss = loss_cls(actor)
        out = loss(td)
        out.get("loss_objective").backward()
        assert torch.isfinite(out.get("loss_objective"))

    def test_grpo_kl_to_ref(self, lm, tok):
        td = self._fake_rollout(lm, tok)
        MCAdvantage(grpo_size=2)(td)
        td.set("ref_log_probs", td.get("log_probs").clone() - 0.1)
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        loss = GRPOLoss(actor, kl_to_ref_coeff=0.1)
        out = loss(td)
        assert "loss_kl_to_ref" in out

    def test_sft(self, lm, tok):
        td = self._fake_rollout(lm, tok)
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        out = SFTLoss(actor)(td)
        out.get("loss_sft").backward()
        assert torch.isfinite(out.get("loss_sft"))


class TestEndToEndGRPO:
    """Full RLHF loop: PromptDataset → ChatEnv → LLMCollector (generate)
    → MCAdvantage → GRPOLoss → Adam step.  Reference shape: pytorch/rl
    sota-implementations/grpo (collector + RB + loss wiring)."""

    def test_grpo_training_loop(self, lm, tok):
        from rl_amd.data.llm import PromptDataset

        prompts = PromptDataset(["2+2?", "3+3?", "4+4?", "5+5?"], repeat=True)
        # deterministic length-based reward: longer answers score higher
        env = ChatEnv(iter(prompts), reward_fn=lambda h: float(len(h.last_content)))
        policy = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=4)
        col = LLMCollector(env, policy, dialog_turns_per_batch=4, total_dialog_turns=8)
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        loss_mod = GRPOLoss(actor)
        optim = torch.optim.Adam(lm.parameters(), lr=1e-4)
        w0 = [p.detach().clone() for p in lm.parameters()]
        n_updates = 0
        for batch in col:
            batch = batch.reshape(-1)
            # strip the generated assistant turn for log-prob recompute
            hist = batch.get_non_tensor("history")
            MCAdvantage(grpo_size=2)(batch)
            out = loss_mod(batch)
            loss = out.get("loss_objective")
            assert torch.isfinite(loss)
            optim.zero_grad()
            loss.backward()
            optim.step()
            n_updates += 1
        assert n_updates == 2
        changed = any(
            not torch.equal(a, b) for a, b in zip(w0, lm.parameters())
        )
        assert changed, "GRPO loop did not update the policy weights"


class TestDistillation:
    def test_distill_zero_when_matching(self, lm, tok):
        from rl_amd.objectives import DistillationLoss

        # build a rollout and use the SAME model as teacher: KL ~ 0
        gen = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=4)
        h = History.from_text(["q"] * 2)
        td = TensorDict({}, batch_size=[2])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td = gen(td)
        hist = td.get_non_tensor("history")
        td.set_non_tensor("history", {"roles": [r[:-1] for r in hist["roles"]], "contents": [c[:-1] for c in hist["contents"]]})
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        scored = actor(td.clone(False))
        td.set("ref_log_probs", scored.get("log_probs").detach())
        loss = DistillationLoss(actor)
        out = loss(td)
        assert out.get("loss_distill").abs() < 1e-4  # same model → KL ≈ 0
        out.get("loss_distill").backward()

    def test_distill_nonzero_and_directional(self, lm, tok):
        from rl_amd.objectives import DistillationLoss

        gen = TransformersWrapper(lm, tokenizer=tok, generate=True, max_new_tokens=4)
        h = History.from_text(["q"] * 2)
        td = TensorDict({}, batch_size=[2])
        td.set_non_tensor("history", {"roles": h.roles, "contents": h.contents})
        td = gen(td)
        hist = td.get_non_tensor("history")
        td.set_non_tensor("history", {"roles": [r[:-1] for r in hist["roles"]], "contents": [c[:-1] for c in hist["contents"]]})
        actor = TransformersWrapper(lm, tokenizer=tok, generate=False)
        scored = actor(td.clone(False))
        td.set("ref_log_probs", scored.get("log_probs").detach() - 0.5)  # shifted teacher
        for d in ("reverse", "forward"):
            out = DistillationLoss(actor, kl_direction=d)(td.clone(False))
            assert torch.isfinite(out.get("loss_distill"))
            assert out.get("loss_distill") > 0


class TestRLHFDataUtils:
    def test_tokenizers_and_prompt_data(self, tok):
        from rl_amd.data import PromptData, PromptTensorDictTokenizer, TensorDictTokenizer

        t = TensorDictTokenizer(tok, max_length=16)
        td = t(["hello world", "hi"])
        assert td["input_ids"].shape == (2, 16)
        pt = PromptTensorDictTokenizer(tok, max_length=16)
        td2 = pt(["abc"])
        assert td2["prompt_rindex"].item() == int(td2["attention_mask"].sum())
        pd = PromptData.from_tensordict(td2)
        rt = pd.to_tensordict()
        assert torch.equal(rt["input_ids"], td2["input_ids"])

    def test_rollout_from_model(self, lm, tok):
        from rl_amd.data import RolloutFromModel
        from rl_amd.modules import RewardModel
        from rl_amd.tensordict import TensorDict

        torch.manual_seed(0)
        rm = RewardModel(model=make_tiny_lm())
        roll = RolloutFromModel(lm, lm, rm, max_new_tokens=4, kl_coef=0.1)
        ids = torch.randint(1, 250, (2, 6))
        batch = TensorDict({"input_ids": ids, "attention_mask": torch.ones_like(ids)}, batch_size=[2])
        td = roll.rollout_from_data(batch)
        assert td.batch_size == torch.Size([2, 4])
        assert td["next", "done"][:, -1].all() and not td["next", "done"][:, :-1].any()
        # same model as reference → KL term ~0 → reward = end score at last step
        assert torch.allclose(
            td["next", "reward"][:, :-1], torch.zeros(2, 3, 1), atol=1e-4
        )

    def test_topk_selector(self):
        from rl_amd.data import TopKRewardSelector
        from rl_amd.tensordict import TensorDict

        n, g = 8, 4
        rewards = torch.arange(n, dtype=torch.float32).reshape(n, 1, 1)
        done = torch.ones(n, 1, 1, dtype=torch.bool)
        td = TensorDict(
            {"x": torch.arange(n).float(), "next": {"reward": rewards, "done": done}},
            batch_size=[n],
        )
        out = TopKRewardSelector(k=2, group_size=g)(td)
        assert sorted(out["x"].tolist()) == [2.0, 3.0, 6.0, 7.0]
