"""Property-based tests (hypothesis): TensorDict indexing must agree
with plain dict-of-tensors semantics; value scans must match their
sequential oracles on arbitrary shapes; codecs must round-trip."""
import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from rl_amd.tensordict import TensorDict


@st.composite
def batch_and_index(draw):
    b0 = draw(st.integers(2, 6))
    b1 = draw(st.integers(2, 5))
    kind = draw(st.sampled_from(["int", "slice", "bool", "tensor", "ellipsis"]))
    if kind == "int":
        idx = draw(st.integers(0, b0 - 1))
    elif kind == "slice":
        lo = draw(st.integers(0, b0 - 1))
        hi = draw(st.integers(lo + 1, b0))
        idx = slice(lo, hi)
    elif kind == "bool":
        idx = torch.rand(b0) > 0.5
    elif kind == "tensor":
        idx = torch.randint(0, b0, (draw(st.integers(1, 4)),))
    else:
        idx = Ellipsis
    return (b0, b1), idx


class TestTensorDictProperties:
    @settings(max_examples=50, deadline=None)
    @given(data=batch_and_index())
    def test_indexing_matches_tensor_semantics(self, data):
        (b0, b1), idx = data
        a = torch.randn(b0, b1, 3)
        c = torch.randn(b0, b1)
        td = TensorDict({"a": a, "nested": {"c": c}}, batch_size=[b0, b1])
        sub = td[idx]
        assert torch.equal(sub.get("a"), a[idx])
        assert torch.equal(sub.get(("nested", "c")), c[idx])
        assert sub.batch_size == a[idx].shape[:2] or sub.batch_size == a[idx].shape[: len(sub.batch_size)]

    @settings(max_examples=30, deadline=None)
    @given(
        b=st.integers(1, 5),
        t=st.integers(1, 6),
        dim=st.integers(0, 1),
    )
    def test_stack_unbind_roundtrip(self, b, t, dim):
        tds = [
            TensorDict({"x": torch.randn(b, t, 2), "n": {"y": torch.randn(b, t)}},
                       batch_size=[b, t])
            for _ in range(3)
        ]
        from rl_amd.tensordict import stack

        s = stack(tds, dim)
        assert s.batch_size[dim] == 3
        for i, td in enumerate(tds):
            back = s[(slice(None),) * dim + (i,)]
            assert torch.equal(back.get("x"), td.get("x"))

    @settings(max_examples=30, deadline=None)
    @given(b0=st.integers(1, 4), b1=st.integers(1, 4))
    def test_reshape_view_consistency(self, b0, b1):
        td = TensorDict({"x": torch.arange(b0 * b1 * 2).reshape(b0, b1, 2).float()},
                        batch_size=[b0, b1])
        flat = td.reshape(-1)
        assert flat.batch_size == torch.Size([b0 * b1])
        assert torch.equal(flat.get("x"), td.get("x").reshape(b0 * b1, 2))


class TestScanProperties:
    @settings(max_examples=25, deadline=None)
    @given(
        b=st.integers(1, 6),
        t=st.integers(1, 32),
        gamma=st.floats(0.5, 0.999),
        lmbda=st.floats(0.5, 1.0),
        p_done=st.floats(0.0, 0.5),
    )
    def test_vec_gae_matches_sequential(self, b, t, gamma, lmbda, p_done):
        from rl_amd.objectives.value import functional as F

        torch.manual_seed(0)
        value = torch.randn(b, t, 1)
        next_value = torch.randn(b, t, 1)
        reward = torch.randn(b, t, 1)
        done = torch.rand(b, t, 1) < p_done
        adv_v, tgt_v = F.vec_generalized_advantage_estimate(
            gamma, lmbda, value, next_value, reward, done, done
        )
        adv_s, tgt_s = F.generalized_advantage_estimate(
            gamma, lmbda, value, next_value, reward, done, done
        )
        assert torch.allclose(adv_v, adv_s, atol=1e-4), (adv_v - adv_s).abs().max()
        assert torch.allclose(tgt_v, tgt_s, atol=1e-4)


class TestCodecProperties:
    @settings(max_examples=40, deadline=None)
    @given(x=st.floats(-50.0, 50.0))
    def test_two_hot_roundtrip(self, x):
        from rl_amd.modules.functional import (
            default_bins, symexp, symlog, two_hot_encode,
        )

        bins = default_bins(255)
        enc = two_hot_encode(symlog(torch.tensor([x])), bins)
        dec = symexp((enc * bins).sum(-1)).item()
        assert abs(dec - x) <= 0.05 + 0.1 * abs(x)

    @settings(max_examples=40, deadline=None)
    @given(n_bins=st.integers(8, 512), v=st.floats(-0.999, 0.999))
    def test_action_tokenizer_roundtrip(self, n_bins, v):
        from rl_amd.data.vla import UniformActionTokenizer

        tok = UniformActionTokenizer(n_bins, -1.0, 1.0)
        a = torch.tensor([[v]])
        rt = tok.decode(tok.encode(a))
        assert (rt - a).abs().max() <= 1.0 / n_bins + 1e-6
