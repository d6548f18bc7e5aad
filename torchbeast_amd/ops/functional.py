"""Autograd wrappers for the fused CDNA4 kernels, with eager oracles.

Every op has two implementations:
- the HIP one (extension `_tbops`, gfx950 kernels), used for CUDA tensors;
- an eager PyTorch one, used on CPU and as the fp32 numerics oracle in tests.

On a GPU machine the eager path is NOT silently substituted: if the
extension lacks a kernel the op raises (TBAMD_ALLOW_EAGER=1 overrides).
"""

import os

import torch
import torch.nn.functional as F

import torchbeast_amd.ops as ops_mod


def _ext_for(tensor, symbol):
    """Return the extension if `tensor` is on GPU and `symbol` exists."""
    if not tensor.is_cuda:
        return None
    ext = ops_mod.require_ext() if not os.environ.get("TBAMD_ALLOW_EAGER") else ops_mod._load()
    if ext is not None and hasattr(ext, symbol):
        return ext
    if os.environ.get("TBAMD_ALLOW_EAGER"):
        return None
    raise RuntimeError(
        f"HIP kernel '{symbol}' missing from _tbops on a GPU tensor "
        "(rebuild the extension, or set TBAMD_ALLOW_EAGER=1)"
    )


# ---------------------------------------------------------------------------
# V-trace (fused): action log-probs + rho/c clip + reverse scan + advantages.
# ---------------------------------------------------------------------------


def vtrace_from_logits(
    behavior_policy_logits,
    target_policy_logits,
    actions,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    from torchbeast_amd.core import vtrace as pyvtrace

    ext = _ext_for(target_policy_logits, "vtrace_from_logits")
    if ext is None:
        # Eager path (also the oracle): delegate to the pure-PyTorch module.
        target_lp = pyvtrace.action_log_probs(target_policy_logits, actions)
        behavior_lp = pyvtrace.action_log_probs(behavior_policy_logits, actions)
        log_rhos = target_lp - behavior_lp
        core = pyvtrace.from_importance_weights(
            log_rhos,
            discounts,
            rewards,
            values,
            bootstrap_value,
            clip_rho_threshold,
            clip_pg_rho_threshold,
        )
        return pyvtrace.VTraceFromLogitsReturns(
            vs=core.vs,
            pg_advantages=core.pg_advantages,
            log_rhos=log_rhos,
            behavior_action_log_probs=behavior_lp,
            target_action_log_probs=target_lp,
        )

    with torch.no_grad():
        vs, pg_adv, log_rhos, blp, tlp = ext.vtrace_from_logits(
            behavior_policy_logits.detach().float().contiguous(),
            target_policy_logits.detach().float().contiguous(),
            actions.contiguous(),
            discounts.detach().float().contiguous(),
            rewards.detach().float().contiguous(),
            values.detach().float().contiguous(),
            bootstrap_value.detach().float().contiguous(),
            float(clip_rho_threshold),
            float(clip_pg_rho_threshold),
        )
    return pyvtrace.VTraceFromLogitsReturns(
        vs=vs,
        pg_advantages=pg_adv,
        log_rhos=log_rhos,
        behavior_action_log_probs=blp,
        target_action_log_probs=tlp,
    )


# ---------------------------------------------------------------------------
# Fused IMPALA loss: pg + baseline + entropy, with analytic gradients.
# ---------------------------------------------------------------------------


class _FusedImpalaLoss(torch.autograd.Function):
    """Forward computes the three scalar losses AND the gradients w.r.t.
    (logits, baseline) in one kernel pass; backward just scales the saved
    gradients by the incoming scalar grads. grads:
      d pg/d logits       = (softmax - onehot(a)) * pg_adv        (adv detached)
      d entropy/d logits  = softmax * (logp - sum(p*logp))
      d baseline/d bl     = baseline - vs
    """

    @staticmethod
    def forward(ctx, logits, baseline, actions, pg_advantages, vs):
        ext = _ext_for(logits, "fused_impala_loss_fwd")
        (
            pg_loss,
            baseline_loss,
            entropy_loss,
            d_logits_pg,
            d_logits_ent,
            d_baseline,
        ) = ext.fused_impala_loss_fwd(
            logits.detach().float().contiguous(),
            baseline.detach().float().contiguous(),
            actions.contiguous(),
            pg_advantages.detach().float().contiguous(),
            vs.detach().float().contiguous(),
        )
        ctx.save_for_backward(d_logits_pg, d_logits_ent, d_baseline)
        return pg_loss, baseline_loss, entropy_loss

    @staticmethod
    def backward(ctx, g_pg, g_bl, g_ent):
        d_logits_pg, d_logits_ent, d_baseline = ctx.saved_tensors
        grad_logits = g_pg * d_logits_pg + g_ent * d_logits_ent
        grad_baseline = g_bl * d_baseline
        return grad_logits, grad_baseline, None, None, None


def fused_impala_loss(logits, baseline, actions, pg_advantages, vs):
    """Return (pg_loss, baseline_loss, entropy_loss) scalars.

    baseline_loss here is 0.5*sum((vs - baseline)^2) — the caller applies the
    cost weights.
    """
    if logits.is_cuda and _ext_for(logits, "fused_impala_loss_fwd") is not None:
        return _FusedImpalaLoss.apply(logits, baseline, actions, pg_advantages, vs)

    from torchbeast_amd.core import losses

    pg_loss = losses.compute_policy_gradient_loss(logits, actions, pg_advantages)
    baseline_loss = losses.compute_baseline_loss(vs - baseline)
    entropy_loss = losses.compute_entropy_loss(logits)
    return pg_loss, baseline_loss, entropy_loss


# ---------------------------------------------------------------------------
# Done-masked LSTM unroll over T (multi-layer).
# ---------------------------------------------------------------------------


class _LstmUnroll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, notdone, h0, c0, *flat_weights):
        ext = _ext_for(x, "lstm_unroll_fwd")
        num_layers = len(flat_weights) // 4
        results = ext.lstm_unroll_fwd(
            x.contiguous(),
            notdone.contiguous(),
            h0.contiguous(),
            c0.contiguous(),
            [w.contiguous() for w in flat_weights],
        )
        out, hT, cT = results[:3]
        stash = results[3:]  # Per layer: input, gates, hm, cm, c.
        ctx.save_for_backward(notdone, h0, c0, *flat_weights, *stash)
        ctx.num_layers = num_layers
        return out, hT, cT

    @staticmethod
    def backward(ctx, d_out, d_hT, d_cT):
        saved = ctx.saved_tensors
        notdone, h0, c0 = saved[:3]
        nw = ctx.num_layers * 4
        flat_weights = list(saved[3 : 3 + nw])
        stash = list(saved[3 + nw :])
        ext = ops_mod.require_ext()
        grads = ext.lstm_unroll_bwd(
            notdone,
            h0,
            c0,
            flat_weights,
            stash,
            d_out.contiguous(),
            d_hT.contiguous(),
            d_cT.contiguous(),
        )
        d_x, d_h0, d_c0 = grads[:3]
        d_weights = grads[3:]
        return (d_x, None, d_h0, d_c0, *d_weights)


def lstm_unroll(core, x, notdone, state):
    """Run `core` (an nn.LSTM) over x [T,B,I] with per-step done masking.

    state is (h, c), each [L, B, H]. Mirrors the model-level Python loop:
        state <- notdone_t * state;  out_t, state <- lstm(x_t, state)
    Returns (output [T,B,H], (hT, cT)).
    """
    if x.is_cuda and _ext_for(x, "lstm_unroll_fwd") is not None:
        flat = []
        for layer in range(core.num_layers):
            flat += [
                getattr(core, f"weight_ih_l{layer}"),
                getattr(core, f"weight_hh_l{layer}"),
                getattr(core, f"bias_ih_l{layer}"),
                getattr(core, f"bias_hh_l{layer}"),
            ]
        out, hT, cT = _LstmUnroll.apply(x, notdone, state[0], state[1], *flat)
        return out, (hT, cT)

    outputs = []
    for xt, nd in zip(x.unbind(), notdone.unbind()):
        nd = nd.view(1, -1, 1)
        state = tuple(nd * s for s in state)
        out, state = core(xt.unsqueeze(0), state)
        outputs.append(out)
    return torch.cat(outputs), state


# ---------------------------------------------------------------------------
# Fused grad-clip + RMSProp (+ optional LR from a lambda schedule).
# ---------------------------------------------------------------------------


def rmsprop_step(param, grad, square_avg, lr, alpha, eps, clip_norm=None,
                 lr_tensor=None):
    """One RMSProp step over FLAT tensors, fused with global-norm clipping.

    All of param/grad/square_avg are 1-D views over the whole model (see
    parallel.flat_params). Returns the pre-clip gradient norm (0-dim tensor).
    """
    ext = _ext_for(param, "rmsprop_step") if param.is_cuda else None
    if ext is not None:
        return ext.rmsprop_step(
            param, grad, square_avg, float(lr), float(alpha), float(eps),
            float(clip_norm) if clip_norm is not None else -1.0,
            lr_tensor,
        )

    if lr_tensor is not None:
        lr = float(lr_tensor.item())
    total_norm = grad.norm(2)
    if clip_norm is not None:
        coef = clip_norm / (total_norm + 1e-6)
        if coef < 1:
            grad = grad.mul(coef)
    square_avg.mul_(alpha).addcmul_(grad, grad, value=1 - alpha)
    avg = square_avg.sqrt().add_(eps)
    param.addcdiv_(grad, avg, value=-lr)
    return total_norm


# ---------------------------------------------------------------------------
# Fused AtariNet conv trunk (u8 frames -> flat conv features).
# ---------------------------------------------------------------------------


def _pack_trunk_weights(w1, w2, w3):
    """bf16 operand layouts the MFMA trunk consumes: conv1 keeps the native
    (c,ky,kx) flatten; conv2/3 are (ky,kx,c)-major for NHWC activations."""
    w1p = w1.detach().reshape(w1.shape[0], -1).to(torch.bfloat16).contiguous()
    w2p = (w2.detach().permute(0, 2, 3, 1).reshape(w2.shape[0], -1)
           .to(torch.bfloat16).contiguous())
    w3p = (w3.detach().permute(0, 2, 3, 1).reshape(w3.shape[0], -1)
           .to(torch.bfloat16).contiguous())
    return w1p, w2p, w3p


class _AtariTrunkMfma(torch.autograd.Function):
    """bf16 MFMA implicit-GEMM trunk (conv_mfma.hip): hand-written forward,
    dgrad and wgrad kernels; fp32 accumulation throughout. Replaces the
    reference's stock conv stack (torchbeast/monobeast.py:552-559,586-589)
    on the learner path."""

    @staticmethod
    def forward(ctx, frames, w1, b1, w2, b2, w3, b3):
        ext = _ext_for(frames, "conv_trunk_fwd")
        w1p, w2p, w3p = _pack_trunk_weights(w1, w2, w3)
        out3, a1, a2 = ext.conv_trunk_fwd(
            frames, w1p, b1.detach().contiguous(), w2p,
            b2.detach().contiguous(), w3p, b3.detach().contiguous(), True
        )
        ctx.save_for_backward(frames, w2, w3, a1, a2, out3)
        return out3

    @staticmethod
    def backward(ctx, d_out3):
        frames, w2, w3, a1, a2, out3 = ctx.saved_tensors
        ext = ops_mod.require_ext()
        d3m = ext.conv_trunk_mask_d3(d_out3.contiguous().float(), out3)
        # dgrad = stride-1 correlation with rotated weights, [ci][ky,kx,co].
        w3r = (w3.detach().flip(2, 3).permute(1, 2, 3, 0)
               .reshape(w3.shape[1], -1).to(torch.bfloat16).contiguous())
        w2r = (w2.detach().flip(2, 3).permute(1, 2, 3, 0)
               .reshape(w2.shape[1], -1).to(torch.bfloat16).contiguous())
        d2 = ext.conv_trunk_dgrad3(d3m, w3r, a2)
        d1 = ext.conv_trunk_dgrad2(d2, w2r, a1)
        dw3p, db3 = ext.conv_trunk_wgrad3(a2, d3m)
        dw2p, db2 = ext.conv_trunk_wgrad2(a1, d2)
        dw1p, db1 = ext.conv_trunk_wgrad1(frames, d1)

        # Kernel dW layout is [ky][co][kx*ci+c] ([ky][co][c*kw+kx] for
        # conv1, K-columns padded to 16); permute back to [co][ci][kh][kw].
        def unpack(dwp, wshape, c_major):
            co, ci, kh, kw = wshape
            kwcp = dwp.shape[-1]
            dwp = dwp.view(kh, co, kwcp)[..., :kw * ci]
            if c_major:  # conv1: (c, kx) minor order
                return (dwp.reshape(kh, co, ci, kw)
                        .permute(1, 2, 0, 3).contiguous())
            return (dwp.reshape(kh, co, kw, ci)
                    .permute(1, 3, 0, 2).contiguous())

        w1_shape = (d1.shape[-1], frames.shape[1], 8, 8)
        dw3 = unpack(dw3p, w3.shape, c_major=False)
        dw2 = unpack(dw2p, w2.shape, c_major=False)
        dw1 = unpack(dw1p, w1_shape, c_major=True)
        return None, dw1, db1, dw2, db2, dw3, db3


def atari_trunk(frames, conv1, conv2, conv3):
    """Fused u8-frame conv trunk for GPU [N,C,H,W] u8 frames.

    Dispatch (both paths are hand-written gfx950 kernels):
    - training or large batches: bf16 MFMA implicit-GEMM kernels
      (conv_mfma.hip) — covers the full learner batch T*B;
    - small no-grad batches (inference service): the single-launch
      per-sample fused trunk (atari_trunk.hip).
    Returns flat post-relu conv3 features [N, 64*H3*W3] (NCHW flatten
    order), or None when no kernel applies (CPU tensors, unsupported
    geometry) — caller falls back to eager convs.
    """
    if not frames.is_cuda or frames.dtype != torch.uint8:
        return None
    force = os.environ.get("TBAMD_TRUNK")  # mfma | valu | lib (benchmarking)
    if force == "lib":
        return None
    mfma_ok = tuple(frames.shape[1:]) in ((4, 84, 84), (3, 210, 160)) \
        and force != "valu"
    grad = torch.is_grad_enabled() and (
        conv1.weight.requires_grad or frames.requires_grad
    )
    if mfma_ok and (grad or frames.shape[0] > 384 or force == "mfma"):
        ext = _ext_for(frames, "conv_trunk_fwd")
        args = (frames.contiguous(), conv1.weight, conv1.bias, conv2.weight,
                conv2.bias, conv3.weight, conv3.bias)
        if grad:
            return _AtariTrunkMfma.apply(*args)
        w1p, w2p, w3p = _pack_trunk_weights(
            conv1.weight, conv2.weight, conv3.weight)
        (out3,) = ext.conv_trunk_fwd(
            frames.contiguous(), w1p, conv1.bias.detach().contiguous(), w2p,
            conv2.bias.detach().contiguous(), w3p,
            conv3.bias.detach().contiguous(), False)
        return out3
    if grad:
        return None  # unsupported geometry: eager autograd convs
    ext = _ext_for(frames, "atari_trunk_fwd")
    if ext is None or not ext.atari_trunk_supported(*frames.shape[1:]):
        return None
    (out3,) = ext.atari_trunk_fwd(
        frames.contiguous(), conv1.weight, conv1.bias, conv2.weight,
        conv2.bias, conv3.weight, conv3.bias, False)
    return out3


# ---------------------------------------------------------------------------
# Deep-ResNet 3x3 s1 p1 conv on the same MFMA implicit-GEMM template
# (activations stay bf16 channels_last through the trunk; see
# models/resnet.py, which mirrors the reference deep net of
# polybeast_learner.py:134-266).
# ---------------------------------------------------------------------------


def _pack_resnet_weight(w, rotate=False):
    """[CO,CI,3,3] fp32 -> bf16 [rows, KP] (ky,kx,c)-major, K zero-padded to
    a multiple of 32 (the kernel's MFMA K-step). rotate=True builds the
    dgrad operand [CI, ky,kx,CO] with spatially flipped taps."""
    if rotate:
        w = w.detach().flip(2, 3).permute(1, 2, 3, 0)
    else:
        w = w.detach().permute(0, 2, 3, 1)
    rows = w.shape[0]
    flat = w.reshape(rows, -1).to(torch.bfloat16)
    k = flat.shape[1]
    kp = (k + 31) // 32 * 32
    if kp != k:
        flat = torch.cat([flat, flat.new_zeros(rows, kp - k)], dim=1)
    return flat.contiguous()


class _ResNetConv3x3(torch.autograd.Function):
    """3x3 s1 p1 conv over bf16 channels_last activations, MFMA fwd/dgrad
    (conv_mfma.hip conv_nhwc_kernel MODE 5/3) + transposed-GEMM wgrad."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        x = x.contiguous(memory_format=torch.channels_last)
        ci, hw, co = x.shape[1], x.shape[2], weight.shape[0]
        ext = _ext_for(x, "resnet_conv")
        wp = _pack_resnet_weight(weight)
        out = ext.resnet_conv(x, wp, bias.detach().float().contiguous(),
                              ci, hw, co, True)
        ctx.save_for_backward(x, weight)
        ctx.geom = (ci, hw, co)
        # NHWC-dense result presented as logical NCHW (channels_last view).
        return out.permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        ci, hw, co = ctx.geom
        ext = ops_mod.require_ext()
        dyn = (dy.to(torch.bfloat16)
               .contiguous(memory_format=torch.channels_last))
        wr = _pack_resnet_weight(weight, rotate=True)
        dx = ext.resnet_conv(dyn, wr, dyn.new_empty(0).float(),
                             co, hw, ci, False).permute(0, 3, 1, 2)
        dwp, db = ext.resnet_conv_wgrad(x, dyn, ci, hw, co)
        # [ky][co][kx*ci+c] (K-cols padded to 16) -> [co][ci][kh][kw].
        dw = (dwp[:, :, :3 * ci].view(3, co, 3, ci)
              .permute(1, 3, 0, 2).contiguous())
        return dx, dw, db


def resnet_conv3x3(conv, x):
    """Apply an nn.Conv2d(3x3, s1, p1) via the MFMA kernel; x is bf16
    channels_last on GPU with a supported geometry (checked by caller)."""
    return _ResNetConv3x3.apply(x, conv.weight, conv.bias)


class _ResNetFirstConv(torch.autograd.Function):
    """First deep-net conv (obs channels -> 16 @84): input channels are
    zero-padded to 8 so the K-runs fill MFMA A-fragments. Frames carry no
    grad, so backward is wgrad-only (the zero-channel taps are sliced
    off). Replaces MIOpen's fp32 NCHW wrw, which falls back to a naive
    double-accumulation kernel on this geometry."""

    @staticmethod
    def forward(ctx, x8, weight, bias):
        co, ci = weight.shape[0], weight.shape[1]
        w8 = torch.cat(
            [weight.detach(),
             weight.new_zeros(co, 8 - ci, *weight.shape[2:])], dim=1)
        ext = _ext_for(x8, "resnet_conv")
        out = ext.resnet_conv(x8, _pack_resnet_weight(w8),
                              bias.detach().float().contiguous(),
                              8, x8.shape[2], co, True)
        ctx.save_for_backward(x8)
        ctx.ci = ci
        return out.permute(0, 3, 1, 2)

    @staticmethod
    def backward(ctx, dy):
        (x8,) = ctx.saved_tensors
        ext = ops_mod.require_ext()
        dyn = (dy.to(torch.bfloat16)
               .contiguous(memory_format=torch.channels_last))
        co = dyn.shape[1]
        dwp, db = ext.resnet_conv_wgrad(x8, dyn, 8, x8.shape[2], co)
        dw = (dwp[:, :, :24].view(3, co, 3, 8).permute(1, 3, 0, 2)
              [:, :ctx.ci].contiguous())
        return None, dw, db


def resnet_first_conv(conv, x):
    """x: fp32 [N, C<=8, 84, 84] frames (no grad); returns bf16
    channels_last conv output."""
    n, c = x.shape[0], x.shape[1]
    x8 = torch.empty((n, 8, *x.shape[2:]), dtype=torch.bfloat16,
                     device=x.device, memory_format=torch.channels_last)
    x8.zero_()
    x8[:, :c] = x
    return _ResNetFirstConv.apply(x8, conv.weight, conv.bias)


def resnet_conv3x3_supported(ci, hw, co):
    try:
        import torchbeast_amd.ops as _ops
        ext = _ops.require_ext()
    except Exception:  # noqa: BLE001
        return False
    return bool(ext.resnet_conv_supported(ci, hw, co))


# ---------------------------------------------------------------------------
# Policy sampling: softmax + multinomial (train) / argmax (eval).
# ---------------------------------------------------------------------------


def policy_sample(policy_logits, greedy=False, generator=None):
    """Sample actions from categorical logits [N, A] -> [N] int64."""
    if policy_logits.is_cuda:
        ext = _ext_for(policy_logits, "policy_sample")
        if ext is not None:
            seed = int(torch.randint(0, 2**62, (1,)).item()) if not greedy else 0
            return ext.policy_sample(
                policy_logits.detach().float().contiguous(), bool(greedy), seed
            )
    if greedy:
        return torch.argmax(policy_logits, dim=-1)
    return torch.multinomial(
        F.softmax(policy_logits, dim=-1), num_samples=1, generator=generator
    ).squeeze(-1)
