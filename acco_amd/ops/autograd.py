"""torch.autograd.Function wrappers around the gfx950 HIP kernels.

Only imported on the GPU path (ops/__init__ dispatch); each Function pairs a
hand-written forward kernel with its hand-written backward kernel, with
fp32 row statistics saved between them. Numerics tests compare each against
the fp32 torch reference (tests/test_gpu_kernels.py).
"""

from __future__ import annotations

import torch

from acco_amd import ops


class SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        gate = gate.contiguous()
        up = up.contiguous()
        ctx.save_for_backward(gate, up)
        return ops.hip_ext().swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, dout):
        gate, up = ctx.saved_tensors
        dg, du = ops.hip_ext().swiglu_bwd(dout.contiguous(), gate, up)
        return dg, du


class SwiGLUPackedFn(torch.autograd.Function):
    """SwiGLU over the packed [.., 2I] fused gate_up output: no
    split/contiguous forward, single dgu write backward (no cat)."""

    @staticmethod
    def forward(ctx, gu):
        gu = gu.contiguous()
        ctx.save_for_backward(gu)
        return ops.hip_ext().swiglu_packed_fwd(gu)

    @staticmethod
    def backward(ctx, dout):
        (gu,) = ctx.saved_tensors
        return ops.hip_ext().swiglu_packed_bwd(dout.contiguous(), gu)


class GeluNewFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.save_for_backward(x)
        return ops.hip_ext().gelu_fwd(x)

    @staticmethod
    def backward(ctx, dout):
        (x,) = ctx.saved_tensors
        return ops.hip_ext().gelu_bwd(dout.contiguous(), x)


class RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        x = x.contiguous()
        y, rstd = ops.hip_ext().rmsnorm_fwd(x, weight.contiguous(), eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        none = torch.empty(0, device=x.device, dtype=x.dtype)
        dx, dw = ops.hip_ext().rmsnorm_bwd(dy.contiguous(), x,
                                           weight.contiguous(), rstd, none)
        return dx, dw.to(weight.dtype), None


class AddRMSNormFn(torch.autograd.Function):
    """Fused residual-add + RMSNorm: (y, s) = (rmsnorm(x+res)·w, bf16(x+res)).
    The backward folds the residual branch's grad (ds) into the norm-bwd
    kernel's dx write — the eager add/accumulate kernels at every residual
    site disappear (guide: fuse elementwise work into the producing kernel)."""

    @staticmethod
    def forward(ctx, x, res, weight, eps):
        y, s, rstd = ops.hip_ext().add_rmsnorm_fwd(
            x.contiguous(), res.contiguous(), weight.contiguous(), eps)
        ctx.save_for_backward(s, weight, rstd)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, rstd = ctx.saved_tensors
        dadd = (ds.contiguous() if ds is not None
                else torch.empty(0, device=s.device, dtype=s.dtype))
        dx, dw = ops.hip_ext().rmsnorm_bwd(dy.contiguous(), s,
                                           weight.contiguous(), rstd, dadd)
        # d(x) == d(res): both inputs feed the same sum
        return dx, dx, dw.to(weight.dtype), None


class LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        x = x.contiguous()
        y, mean, rstd = ops.hip_ext().layernorm_fwd(x, weight.contiguous(),
                                                    bias.contiguous(), eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        none = torch.empty(0, device=x.device, dtype=x.dtype)
        dx, dw, db = ops.hip_ext().layernorm_bwd(dy.contiguous(), x,
                                                 weight.contiguous(), mean,
                                                 rstd, none)
        return dx, dw.to(weight.dtype), db.to(weight.dtype), None


class AddLayerNormFn(torch.autograd.Function):
    """Fused residual-add + LayerNorm (GPT-Neo blocks); see AddRMSNormFn."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps):
        y, s, mean, rstd = ops.hip_ext().add_layernorm_fwd(
            x.contiguous(), res.contiguous(), weight.contiguous(),
            bias.contiguous(), eps)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y, s

    @staticmethod
    def backward(ctx, dy, ds):
        s, weight, mean, rstd = ctx.saved_tensors
        dadd = (ds.contiguous() if ds is not None
                else torch.empty(0, device=s.device, dtype=s.dtype))
        dx, dw, db = ops.hip_ext().layernorm_bwd(dy.contiguous(), s,
                                                 weight.contiguous(), mean,
                                                 rstd, dadd)
        return dx, dx, dw.to(weight.dtype), db.to(weight.dtype), None


class RoPEFn(torch.autograd.Function):
    """q: [B,S,H,D], k: [B,S,Hkv,D]; cos/sin fp32 [S,D] host-precomputed."""

    @staticmethod
    def forward(ctx, q, k, cos, sin):
        cos = cos.float().contiguous()
        sin = sin.float().contiguous()
        ext = ops.hip_ext()
        q2 = ext.rope_fwd(q.contiguous(), cos, sin, False)
        k2 = ext.rope_fwd(k.contiguous(), cos, sin, False)
        ctx.save_for_backward(cos, sin)
        return q2, k2

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        ext = ops.hip_ext()
        dq0 = ext.rope_fwd(dq.contiguous(), cos, sin, True)
        dk0 = ext.rope_fwd(dk.contiguous(), cos, sin, True)
        return dq0, dk0, None, None


class CausalLMLossFn(torch.autograd.Function):
    """Fused shifted CE: logits [B,S,V] bf16, labels [B,S] int64.
    epsilon != 0 adds HF-LabelSmoother semantics inside the same kernel
    pass."""

    @staticmethod
    def forward(ctx, logits, labels, epsilon=0.0):
        logits = logits.contiguous()
        labels = labels.contiguous()
        acc, lse = ops.hip_ext().ce_fwd(logits, labels, float(epsilon))
        ctx.save_for_backward(logits, labels, lse, acc)
        ctx.epsilon = float(epsilon)
        return acc[0] / acc[1].clamp(min=1.0)

    @staticmethod
    def backward(ctx, grad_out):
        logits, labels, lse, acc = ctx.saved_tensors
        ext = ops.hip_ext()
        if grad_out.is_cuda and hasattr(ext, "ce_bwd_dev"):
            # read the upstream grad on-device — float(grad_out) would be a
            # D2H sync stalling the backward launch pipeline every micro
            dlogits = ext.ce_bwd_dev(logits, labels, lse, acc,
                                     grad_out.reshape(1).float(), ctx.epsilon)
        else:
            dlogits = ext.ce_bwd(logits, labels, lse, acc, float(grad_out),
                                 ctx.epsilon)
        return dlogits, None, None


class AttnQKVPackedFn(torch.autograd.Function):
    """The whole attention core over the PACKED fused projection output
    [B, S, (H+2Hkv)·D]: optional RoPE (strided kernel on the q/k sections) +
    flash attention v4 forward; backward emits one packed grad — no
    torch.split forward copies, no cat backward. D=64, S%256==0.
    Section offsets support both packing orders (Llama q|k|v,
    GPT-Neo k|v|q); window>0 = GPT-Neo local attention."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, H, Hkv, D, scale, window, offs):
        ext = ops.hip_ext()
        qkv = qkv.contiguous()
        q_off, k_off, v_off = offs
        if cos is not None:
            cos = cos.float().contiguous()
            sin = sin.float().contiguous()
            roped = torch.empty_like(qkv)
            ext.rope_packed(qkv, roped, cos, sin, q_off, H, D, False)
            ext.rope_packed(qkv, roped, cos, sin, k_off, Hkv, D, False)
            roped[..., v_off:v_off + Hkv * D].copy_(
                qkv[..., v_off:v_off + Hkv * D])
        else:
            roped = qkv
        o, lse = ext.attn_fwd_packed(roped, H, Hkv, D, float(scale),
                                     int(window or 0), q_off, k_off, v_off)
        ctx.save_for_backward(roped, o, lse,
                              cos if cos is not None else torch.empty(0),
                              sin if sin is not None else torch.empty(0))
        ctx.meta = (H, Hkv, D, float(scale), int(window or 0), offs)
        return o                     # [B, S, H*D]

    @staticmethod
    def backward(ctx, dO):
        ext = ops.hip_ext()
        roped, o, lse, cos, sin = ctx.saved_tensors
        H, Hkv, D, scale, window, offs = ctx.meta
        q_off, k_off, v_off = offs
        B, S, W = roped.shape
        dO = dO.contiguous()
        delta = ext.attn_delta(dO.view(B, S, H, D), o.view(B, S, H, D))
        dqkv = torch.empty_like(roped)
        dkq, dvq = ext.attn_bwd_packed(roped, dO, lse, delta, dqkv, H, Hkv,
                                       D, scale, window, q_off, k_off, v_off)
        rep = H // Hkv
        if hasattr(ext, "attn_gqa_reduce"):
            # one pass: group-sum over the rep query heads + bf16 cast +
            # strided write into the packed grad's k|v sections (replaces
            # two dim-3 sums, two casts and two strided copies per layer)
            ext.attn_gqa_reduce(dkq, dvq, dqkv, Hkv, rep, D, k_off, v_off)
        elif rep > 1:
            dk = dkq.view(B, S, Hkv, rep, D).sum(3, dtype=torch.float32)
            dv = dvq.view(B, S, Hkv, rep, D).sum(3, dtype=torch.float32)
            dqkv[..., k_off:k_off + Hkv * D] = dk.reshape(B, S, Hkv * D).bfloat16()
            dqkv[..., v_off:v_off + Hkv * D] = dv.reshape(B, S, Hkv * D).bfloat16()
        else:
            dqkv[..., k_off:k_off + Hkv * D] = dkq.reshape(B, S, Hkv * D)
            dqkv[..., v_off:v_off + Hkv * D] = dvq.reshape(B, S, Hkv * D)
        if cos.numel():
            # inverse rotation in place on the q/k grad sections
            ext.rope_packed(dqkv, dqkv, cos, sin, q_off, H, D, True)
            ext.rope_packed(dqkv, dqkv, cos, sin, k_off, Hkv, D, True)
        return (dqkv, None, None, None, None, None, None, None, None)


class AttentionFn(torch.autograd.Function):
    """Flash-style causal attention, gfx950 MFMA (fwd: online softmax;
    bwd: FA2-style recompute, dq + dkv kernels). [B,S,H,D] layout,
    D ∈ {64,128}, S % 64 == 0 (the dispatch falls back to torch_ref for
    other shapes)."""

    @staticmethod
    def forward(ctx, q, k, v, scale, window):
        import math
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        w = int(window) if window else 0
        q = q.contiguous()
        k = k.contiguous()
        v = v.contiguous()
        o, lse = ops.hip_ext().attn_fwd(q, k, v, float(scale), w)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.scale = float(scale)
        ctx.window = w
        return o

    @staticmethod
    def backward(ctx, dO):
        q, k, v, o, lse = ctx.saved_tensors
        dO = dO.contiguous()
        # Delta[b,h,s] = rowsum(dO ∘ O) in fp32, laid out [B,H,S] (one
        # fused kernel; was a 4-kernel ATen cast/mul/reduce chain)
        delta = ops.hip_ext().attn_delta(dO, o)
        dq, dkq, dvq = ops.hip_ext().attn_bwd(q, k, v, dO, lse, delta,
                                              ctx.scale, ctx.window)
        B, S, H, D = q.shape
        Hkv = k.shape[2]
        if Hkv != H:
            rep = H // Hkv
            dk = dkq.view(B, S, Hkv, rep, D).sum(3, dtype=torch.float32).to(k.dtype)
            dv = dvq.view(B, S, Hkv, rep, D).sum(3, dtype=torch.float32).to(v.dtype)
        else:
            dk, dv = dkq, dvq
        return dq, dk, dv, None, None
