"""Model forward signatures and shapes (ref test strategy:
tests/polybeast_net_test.py)."""

import pytest
import torch

from torchbeast_amd.models import AtariNet, ResNet


def _inputs(T, B, shape=(4, 84, 84)):
    return dict(
        frame=torch.randint(0, 256, (T, B, *shape), dtype=torch.uint8),
        reward=torch.randn(T, B),
        done=torch.zeros(T, B, dtype=torch.bool),
        last_action=torch.zeros(T, B, dtype=torch.int64),
    )


@pytest.mark.parametrize("use_lstm", [False, True])
def test_atari_net_shapes(use_lstm):
    T, B, A = 3, 2, 6
    net = AtariNet((4, 84, 84), A, use_lstm=use_lstm)
    state = net.initial_state(batch_size=B)
    if use_lstm:
        assert len(state) == 2
        assert all(s.shape == (2, B, 519) for s in state)
    else:
        assert state == ()

    out, new_state = net(_inputs(T, B), state)
    assert out["policy_logits"].shape == (T, B, A)
    assert out["baseline"].shape == (T, B)
    assert out["action"].shape == (T, B)
    assert out["action"].dtype == torch.int64
    assert (out["action"] >= 0).all() and (out["action"] < A).all()
    if use_lstm:
        assert all(s.shape == (2, B, 519) for s in new_state)


@pytest.mark.parametrize("use_lstm", [False, True])
def test_resnet_shapes(use_lstm):
    T, B, A = 2, 3, 6
    net = ResNet((4, 84, 84), A, use_lstm=use_lstm)
    state = net.initial_state(batch_size=B)
    (action, policy_logits, baseline), new_state = net(_inputs(T, B), state)
    assert policy_logits.shape == (T, B, A)
    assert baseline.shape == (T, B)
    assert action.shape == (T, B)
    if use_lstm:
        assert all(s.shape == (1, B, 256) for s in new_state)


def test_atari_net_eval_mode_greedy():
    net = AtariNet((4, 84, 84), 6)
    net.eval()
    inputs = _inputs(1, 4)
    out, _ = net(inputs, ())
    expected = torch.argmax(out["policy_logits"], dim=-1)
    torch.testing.assert_close(out["action"], expected)


def test_lstm_done_masking_resets_state():
    """With done=True at every step the LSTM state must be zeroed before each
    step, so two different initial states give identical outputs."""
    T, B = 4, 2
    net = AtariNet((4, 84, 84), 6, use_lstm=True)
    net.eval()
    inputs = _inputs(T, B)
    inputs["done"] = torch.ones(T, B, dtype=torch.bool)

    torch.manual_seed(0)
    state_a = tuple(torch.randn(2, B, 519) for _ in range(2))
    state_b = tuple(torch.randn(2, B, 519) for _ in range(2))
    out_a, _ = net(inputs, state_a)
    out_b, _ = net(inputs, state_b)
    torch.testing.assert_close(out_a["policy_logits"], out_b["policy_logits"])


def test_resnet_trunk_output_3872():
    net = ResNet((4, 84, 84), 6)
    assert net.fc.in_features == 11 * 11 * 32  # 3872, ref polybeast_learner.py:194
