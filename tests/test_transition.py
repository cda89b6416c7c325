import pytest
import torch as t

from machin_amd.frame.transition import Transition, TransitionBase


def make_transition(**kwargs):
    return Transition(
        state={"state": t.zeros(1, 4)},
        action={"action": t.zeros(1, 1, dtype=t.long)},
        next_state={"state": t.ones(1, 4)},
        reward=1.0,
        terminal=False,
        **kwargs,
    )


class TestTransition:
    def test_attrs(self):
        tr = make_transition(extra="x")
        assert tr.major_attr == ["state", "action", "next_state"]
        assert tr.sub_attr == ["reward", "terminal"]
        assert tr.custom_attr == ["extra"]
        assert tr.has_keys(["state", "reward", "extra"])
        assert not tr.has_keys(["nope"])
        assert len(tr) == 6
        assert tr["reward"] == 1.0

    def test_batch_size(self):
        assert make_transition().batch_size == 1
        with pytest.raises(ValueError):
            Transition(
                state={"state": t.zeros(2, 4)},
                action={"action": t.zeros(2, 1)},
                next_state={"state": t.zeros(2, 4)},
                reward=0.0,
                terminal=False,
            )

    def test_validation(self):
        with pytest.raises(ValueError):
            Transition(
                state={"state": "not a tensor"},
                action={"action": t.zeros(1, 1)},
                next_state={"state": t.zeros(1, 4)},
                reward=0.0,
                terminal=False,
            )
        with pytest.raises(ValueError):
            Transition(
                state={"state": t.zeros(1, 4)},
                action={"action": t.zeros(1, 1)},
                next_state={"state": t.zeros(1, 4)},
                reward="bad",
                terminal=False,
            )

    def test_no_new_attrs(self):
        tr = make_transition()
        with pytest.raises(RuntimeError):
            tr.something_new = 1
        with pytest.raises(RuntimeError):
            tr["something_new"] = 1

    def test_to_device(self):
        tr = make_transition()
        tr.to("cpu")
        assert tr.state["state"].device.type == "cpu"

    def test_clone_detach(self):
        x = t.zeros(1, 4, requires_grad=True)
        tr = Transition(
            state={"state": x * 2},
            action={"action": t.zeros(1, 1)},
            next_state={"state": t.zeros(1, 4)},
            reward=0.5,
            terminal=True,
        )
        c = tr.clone()._detach()
        assert not c.state["state"].requires_grad
        c.state["state"] += 1
        assert tr.state["state"].sum() == 0  # deep copy

    def test_tensor_sub_attr(self):
        tr = Transition(
            state={"state": t.zeros(1, 4)},
            action={"action": t.zeros(1, 1)},
            next_state={"state": t.zeros(1, 4)},
            reward=t.tensor([[0.3]]),
            terminal=False,
        )
        assert t.is_tensor(tr.reward)

    def test_custom_base(self):
        tb = TransitionBase(
            major_attr=["m"],
            sub_attr=["s"],
            custom_attr=["c"],
            major_data=[{"k": t.zeros(5, 2)}],
            sub_data=[t.zeros(5, 1)],
            custom_data=[["anything"]],
        )
        assert tb.batch_size == 5
        assert tb.c == ["anything"]
