"""TensorSpec contract tests (reference test strategy: test_specs.py)."""
import pytest
import torch

from rl_amd.data import (
    Binary,
    Bounded,
    Categorical,
    Composite,
    MultiCategorical,
    MultiOneHot,
    NonTensor,
    OneHot,
    Unbounded,
    stack_specs,
)


@pytest.mark.parametrize(
    "spec",
    [
        Unbounded(shape=(3,)),
        Bounded(low=-1, high=1, shape=(4,)),
        Categorical(5),
        OneHot(4),
        MultiOneHot([2, 3]),
        MultiCategorical([3, 4]),
        Binary(3),
    ],
)
class TestSpecContract:
    def test_rand_is_in(self, spec):
        for _ in range(5):
            assert spec.is_in(spec.rand())

    def test_rand_shape(self, spec):
        s = spec.rand((7,))
        assert s.shape[0] == 7

    def test_zero(self, spec):
        z = spec.zero()
        assert z.shape == spec.shape

    def test_clone_eq(self, spec):
        assert spec.clone() == spec

    def test_expand(self, spec):
        e = spec.expand(6, *spec.shape)
        assert e.shape[0] == 6
        assert e.is_in(e.rand())


class TestBounded:
    def test_project(self):
        spec = Bounded(low=-1, high=1, shape=(3,))
        out = spec.project(torch.tensor([5.0, -5.0, 0.0]))
        assert out.tolist() == [1.0, -1.0, 0.0]

    def test_per_element_bounds(self):
        spec = Bounded(low=torch.tensor([0.0, -2.0]), high=torch.tensor([1.0, 2.0]))
        for _ in range(10):
            r = spec.rand()
            assert 0 <= r[0] <= 1 and -2 <= r[1] <= 2

    def test_to_device_noop(self):
        spec = Bounded(low=-1, high=1, shape=(2,))
        assert spec.to("cpu").device == torch.device("cpu")


class TestDiscrete:
    def test_onehot_categorical_conversion(self):
        oh = OneHot(5)
        c = Categorical(5)
        sample = oh.rand()
        idx = oh.to_categorical(sample)
        assert c.is_in(idx)
        back = c.to_one_hot(idx)
        assert (back.bool() == sample.bool()).all()

    def test_categorical_project(self):
        spec = Categorical(3)
        assert spec.project(torch.tensor(7)).item() == 2

    def test_multionehot(self):
        spec = MultiOneHot([2, 3])
        r = spec.rand((4,))
        assert r.shape == (4, 5)
        cats = spec.to_categorical(r)
        assert cats.shape == (4, 2)


class TestComposite:
    def test_nested(self):
        comp = Composite(shape=())
        comp[("a", "b")] = Unbounded(shape=(2,))
        comp["c"] = Categorical(3)
        assert ("a", "b") in comp.keys(True, True)
        td = comp.rand()
        assert td.get(("a", "b")).shape == (2,)

    def test_is_in(self):
        comp = Composite({"x": Bounded(low=0, high=1, shape=(2,))}, shape=())
        assert comp.is_in(comp.rand())

    def test_batched_composite(self):
        comp = Composite({"x": Unbounded(shape=(4, 3))}, shape=(4,))
        td = comp.rand()
        assert td.batch_size == torch.Size([4])
        td2 = comp.rand((5,))
        assert td2.batch_size == torch.Size([5, 4])

    def test_update_select_exclude(self):
        a = Composite({"x": Unbounded(shape=(2,))}, shape=())
        b = Composite({"y": Categorical(2)}, shape=())
        a.update(b)
        assert "y" in a
        sel = a.select("x")
        assert "y" not in sel
        exc = a.exclude("x")
        assert "x" not in exc

    def test_expand(self):
        comp = Composite({"x": Unbounded(shape=(3,))}, shape=())
        e = comp.expand(5)
        assert e["x"].shape == (5, 3)

    def test_stack_specs(self):
        s = stack_specs([Unbounded(shape=(3,)) for _ in range(4)], 0)
        assert s.shape == (4, 3)

    def test_nontensor(self):
        comp = Composite({"meta": NonTensor(example_data="hi")}, shape=())
        td = comp.rand()
        assert td.get_non_tensor("meta") == "hi"
