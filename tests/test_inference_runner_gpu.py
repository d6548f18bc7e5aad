"""C++ InferenceRunner numerics vs the Python model forward (GPU)."""

import threading

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("requires ROCm GPU", allow_module_level=True)

from torchbeast_amd import polybeast_learner as pbl  # noqa: E402
from torchbeast_amd import runtime  # noqa: E402
from torchbeast_amd.models.atari_net import AtariNet  # noqa: E402
from torchbeast_amd.models.resnet import ResNet  # noqa: E402


def _serve_once(model, b, use_lstm):
    batcher = runtime.DynamicBatcher(batch_dim=1, minimum_batch_size=1)
    runner = pbl.make_inference_runner(model, batcher)
    runner.start(1)

    torch.manual_seed(0)
    frame = torch.randint(0, 256, (1, b, 4, 84, 84), dtype=torch.uint8)
    reward = torch.randn(1, b)
    done = torch.rand(1, b) < 0.3
    if use_lstm:
        L, H = model.core.num_layers, model.core.hidden_size
        state = (torch.randn(L, b, H), torch.randn(L, b, H))
    else:
        state = ()

    result = {}

    def call():
        result["out"] = batcher.compute(
            ((frame, reward, done, torch.zeros(1, b, dtype=torch.int32),
              torch.zeros(1, b)), state)
        )

    t = threading.Thread(target=call)
    t.start()
    t.join(60)
    runner.stop()
    assert "out" in result, "runner did not serve the batch"
    (action, logits, baseline), new_state = result["out"]

    # Reference: the Python model on the same inputs.
    model.train()
    with torch.no_grad():
        ref_out, ref_state = model(
            dict(frame=frame.cuda(), reward=reward.cuda(), done=done.cuda()),
            tuple(s.cuda() for s in state),
        )
    ref_out = pbl._as_agent_output(ref_out)
    return (action, logits, baseline, new_state), ref_out, ref_state


@pytest.mark.parametrize("use_lstm", [False, True])
def test_shallow_runner_matches_model(use_lstm):
    model = AtariNet((4, 84, 84), 6, use_lstm=use_lstm,
                     use_last_action=False).cuda()
    (action, logits, baseline, new_state), ref_out, ref_state = _serve_once(
        model, b=5, use_lstm=use_lstm
    )
    # The runner serves the trunk from the bf16 MFMA kernels while the
    # reference model forward here runs fp32; tolerances are set for that
    # operand rounding (behavior-policy logits, not learner math).
    torch.testing.assert_close(logits, ref_out[1].cpu(), rtol=5e-2, atol=5e-3)
    torch.testing.assert_close(baseline, ref_out[2].cpu(), rtol=5e-2,
                               atol=5e-3)
    assert action.shape == (1, 5) and action.dtype == torch.int64
    if use_lstm:
        for s, r in zip(new_state, ref_state):
            torch.testing.assert_close(s, r.cpu(), rtol=5e-2, atol=5e-3)


def test_deep_runner_matches_model():
    model = ResNet((4, 84, 84), 6).cuda()
    (action, logits, baseline, new_state), ref_out, _ = _serve_once(
        model, b=3, use_lstm=False
    )
    # Both sides run the bf16 MFMA trunk, but normalization rounding
    # differs (model: fp32/255 -> bf16; runner: u8 -> bf16, bf16 mul).
    torch.testing.assert_close(logits, ref_out[1].cpu(), rtol=5e-2, atol=5e-3)
    torch.testing.assert_close(baseline, ref_out[2].cpu(), rtol=5e-2,
                               atol=5e-3)
